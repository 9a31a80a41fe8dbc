"""containerd hook-registration installer (tools/install_containerd.py).

The reference's installer (tools/install.sh) swapped host
nvidia-container-toolkit binaries; this one's only job is pointing
containerd's base_runtime_spec at an OCI spec carrying the egpu prestart
hook — covered here against 1.x and 2.x config shapes, idempotency, and
the reuse-existing-spec path.
"""
from __future__ import annotations

import importlib.util
import json
import os
import sys

import tomli

_TOOLS = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "tools")
_spec = importlib.util.spec_from_file_location(
    "install_containerd", os.path.join(_TOOLS, "install_containerd.py"))
ic = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(ic)


CONFIG_1X = """\
version = 2

[plugins."io.containerd.grpc.v1.cri"]
  sandbox_image = "registry.k8s.io/pause:3.9"

[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.runc]
  runtime_type = "io.containerd.runc.v2"

[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.runc.options]
  SystemdCgroup = true
"""

CONFIG_2X = """\
version = 3

[plugins.'io.containerd.cri.v1.runtime'.containerd.runtimes.runc]
  runtime_type = 'io.containerd.runc.v2'

[plugins.'io.containerd.cri.v1.runtime'.containerd.runtimes.runc.options]
  SystemdCgroup = true
"""

CONFIG_NO_OPTIONS = """\
version = 2

[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.runc]
  runtime_type = "io.containerd.runc.v2"
"""


def _run(tmp_path, config_text, argv_extra=()):
    cfg = tmp_path / "config.toml"
    cfg.write_text(config_text)
    spec = tmp_path / "egpu-base.json"
    rc = ic.main(["--config", str(cfg), "--spec", str(spec),
                  "--hook", "/usr/local/bin/egpu-hook", *argv_extra])
    return rc, cfg, spec


def _assert_registered(cfg, spec):
    parsed = tomli.loads(cfg.read_text())
    assert ic.existing_base_spec(parsed) == str(spec)
    body = json.loads(spec.read_text())
    hooks = body["hooks"]["prestart"]
    assert [h["path"] for h in hooks] == ["/usr/local/bin/egpu-hook"]
    assert hooks[0]["args"] == ["egpu-hook", "prestart"]
    # must still be a COMPLETE runtime spec, not a bare hooks fragment
    for key in ("ociVersion", "process", "root", "mounts", "linux"):
        assert key in body, key


def test_install_1x(tmp_path):
    rc, cfg, spec = _run(tmp_path, CONFIG_1X)
    assert rc == 0
    _assert_registered(cfg, spec)
    # untouched settings survive
    parsed = tomli.loads(cfg.read_text())
    opts = parsed["plugins"]["io.containerd.grpc.v1.cri"]["containerd"][
        "runtimes"]["runc"]["options"]
    assert opts["SystemdCgroup"] is True
    assert os.path.exists(str(cfg) + ".egpu-bak")


def test_install_2x(tmp_path):
    rc, cfg, spec = _run(tmp_path, CONFIG_2X)
    assert rc == 0
    _assert_registered(cfg, spec)


def test_install_creates_missing_options_table(tmp_path):
    rc, cfg, spec = _run(tmp_path, CONFIG_NO_OPTIONS)
    assert rc == 0
    _assert_registered(cfg, spec)


def test_idempotent(tmp_path):
    rc, cfg, spec = _run(tmp_path, CONFIG_1X)
    assert rc == 0
    cfg_after = cfg.read_text()
    spec_after = spec.read_text()
    rc = ic.main(["--config", str(cfg), "--spec", str(spec),
                  "--hook", "/usr/local/bin/egpu-hook"])
    assert rc == 0
    assert cfg.read_text() == cfg_after
    assert spec.read_text() == spec_after
    hooks = json.loads(spec.read_text())["hooks"]["prestart"]
    assert len(hooks) == 1


def test_reuses_existing_base_spec(tmp_path):
    """A config that already names a base_runtime_spec keeps it; the hook is
    merged into that spec without touching the TOML."""
    existing = tmp_path / "custom-base.json"
    existing.write_text(json.dumps(
        {"ociVersion": "1.1.0", "process": {"cwd": "/"}, "root": {"path": "rootfs"},
         "mounts": [], "linux": {},
         "hooks": {"prestart": [{"path": "/opt/other-hook"}]}}))
    config = CONFIG_1X.replace(
        "SystemdCgroup = true",
        f'SystemdCgroup = true\n  base_runtime_spec = "{existing}"')
    cfg = tmp_path / "config.toml"
    cfg.write_text(config)
    rc = ic.main(["--config", str(cfg), "--spec", str(tmp_path / "unused.json"),
                  "--hook", "/usr/local/bin/egpu-hook"])
    assert rc == 0
    assert cfg.read_text() == config  # TOML untouched
    hooks = json.loads(existing.read_text())["hooks"]["prestart"]
    assert [h["path"] for h in hooks] == ["/opt/other-hook",
                                          "/usr/local/bin/egpu-hook"]
    assert not (tmp_path / "unused.json").exists()


def test_dry_run_touches_nothing(tmp_path):
    cfg = tmp_path / "config.toml"
    cfg.write_text(CONFIG_1X)
    spec = tmp_path / "egpu-base.json"
    rc = ic.main(["--config", str(cfg), "--spec", str(spec),
                  "--hook", "/usr/local/bin/egpu-hook", "--dry-run"])
    assert rc == 0
    assert cfg.read_text() == CONFIG_1X
    assert not spec.exists()


def test_rejects_config_without_runtime(tmp_path):
    cfg = tmp_path / "config.toml"
    cfg.write_text("version = 2\n")
    rc = ic.main(["--config", str(cfg), "--spec", str(tmp_path / "s.json"),
                  "--hook", "/usr/local/bin/egpu-hook"])
    assert rc == 1
    assert cfg.read_text() == "version = 2\n"


def test_host_root_split_view(tmp_path):
    """Init-container scenario: the TOML records the HOST-view spec path
    while the file is written through the /host mount (the agent-view vs.
    advertise-view split that bit the limits mounts too)."""
    host = tmp_path / "host"
    (host / "etc" / "containerd").mkdir(parents=True)
    cfg = host / "etc" / "containerd" / "config.toml"
    cfg.write_text(CONFIG_1X)
    rc = ic.main(["--config", str(cfg),
                  "--spec", "/etc/containerd/egpu-base.json",
                  "--host-root", str(host),
                  "--hook", "/usr/local/bin/egpu-hook"])
    assert rc == 0
    parsed = tomli.loads(cfg.read_text())
    # TOML carries the host-view path verbatim
    assert ic.existing_base_spec(parsed) == "/etc/containerd/egpu-base.json"
    # ... but the file landed under the mount
    body = json.loads((host / "etc" / "containerd" / "egpu-base.json").read_text())
    assert body["hooks"]["prestart"][0]["path"] == "/usr/local/bin/egpu-hook"
