#!/usr/bin/env python3
"""Register the egpu-hook OCI prestart hook with containerd (ROADMAP item 6).

containerd has no hooks.d: the supported way to add an OCI hook is a
``base_runtime_spec`` — a complete OCI runtime spec JSON the runtime starts
from — configured on the runc runtime's options table. This installer makes
that edit end-to-end and idempotently:

1. Builds (or merges into) the base spec JSON: the containerd default spec
   (from ``ctr oci spec`` when available, else an embedded equivalent) plus
   ``hooks.prestart = [egpu-hook]``.
2. Points ``base_runtime_spec`` at it in /etc/containerd/config.toml —
   unless the config already names one, in which case THAT file gets the
   hook merged instead and the config is left untouched.
3. Validates the patched TOML parses and backs up the original
   (``config.toml.egpu-bak``). Re-running is a no-op.

Works with both containerd 1.x (``io.containerd.grpc.v1.cri``) and 2.x
(``io.containerd.cri.v1.runtime``) table names.

The reference's installer (tools/install.sh) swapped the
host's nvidia-container-toolkit binaries for forks; ROCm needs no toolkit,
so registration is the only host-runtime change this framework makes.
"""
from __future__ import annotations

import argparse
import json
import os
import re
import shutil
import subprocess
import sys

HOOK_DEFAULT = "/usr/local/bin/egpu-hook"

# containerd's default OCI spec (shape of `ctr oci spec`), used when ctr is
# not on PATH. Fields the runtime always overrides per-container (args, env,
# cgroupsPath, ...) keep their defaults here.
DEFAULT_SPEC = {
    "ociVersion": "1.1.0",
    "process": {
        "user": {"uid": 0, "gid": 0},
        "args": ["sh"],
        "env": ["PATH=/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin"],
        "cwd": "/",
        "capabilities": {
            key: [
                "CAP_CHOWN", "CAP_DAC_OVERRIDE", "CAP_FSETID", "CAP_FOWNER",
                "CAP_MKNOD", "CAP_NET_RAW", "CAP_SETGID", "CAP_SETUID",
                "CAP_SETFCAP", "CAP_SETPCAP", "CAP_NET_BIND_SERVICE",
                "CAP_SYS_CHROOT", "CAP_KILL", "CAP_AUDIT_WRITE",
            ]
            for key in ("bounding", "effective", "permitted")
        },
        "rlimits": [{"type": "RLIMIT_NOFILE", "hard": 1024, "soft": 1024}],
        "noNewPrivileges": True,
    },
    "root": {"path": "rootfs"},
    "mounts": [
        {"destination": "/proc", "type": "proc", "source": "proc",
         "options": ["nosuid", "noexec", "nodev"]},
        {"destination": "/dev", "type": "tmpfs", "source": "tmpfs",
         "options": ["nosuid", "strictatime", "mode=755", "size=65536k"]},
        {"destination": "/dev/pts", "type": "devpts", "source": "devpts",
         "options": ["nosuid", "noexec", "newinstance", "ptmxmode=0666",
                     "mode=0620", "gid=5"]},
        {"destination": "/dev/shm", "type": "tmpfs", "source": "shm",
         "options": ["nosuid", "noexec", "nodev", "mode=1777", "size=65536k"]},
        {"destination": "/dev/mqueue", "type": "mqueue", "source": "mqueue",
         "options": ["nosuid", "noexec", "nodev"]},
        {"destination": "/sys", "type": "sysfs", "source": "sysfs",
         "options": ["nosuid", "noexec", "nodev", "ro"]},
        {"destination": "/run", "type": "tmpfs", "source": "tmpfs",
         "options": ["nosuid", "strictatime", "mode=755", "size=65536k"]},
    ],
    "linux": {
        "resources": {"devices": [{"allow": False, "access": "rwm"}]},
        "namespaces": [{"type": "pid"}, {"type": "ipc"}, {"type": "uts"},
                       {"type": "mount"}, {"type": "network"}],
        "maskedPaths": [
            "/proc/acpi", "/proc/asound", "/proc/kcore", "/proc/keys",
            "/proc/latency_stats", "/proc/timer_list", "/proc/timer_stats",
            "/proc/sched_debug", "/sys/firmware", "/sys/devices/virtual/powercap",
            "/proc/scsi",
        ],
        "readonlyPaths": [
            "/proc/bus", "/proc/fs", "/proc/irq", "/proc/sys", "/proc/sysrq-trigger",
        ],
    },
}

OPTIONS_TABLE_RE = re.compile(
    r"^\[plugins\.(?:\"[^\"]*\"|'[^']*'|[\w.-]+)"
    r"\.containerd\.runtimes\.(?P<rt>[\w-]+)\.options\]\s*$",
    re.M,
)
RUNTIME_TABLE_RE = re.compile(
    r"^\[plugins\.(?P<cri>\"[^\"]*\"|'[^']*'|[\w.-]+)"
    r"\.containerd\.runtimes\.(?P<rt>[\w-]+)\]\s*$",
    re.M,
)


def default_spec() -> dict:
    ctr = shutil.which("ctr")
    if ctr:
        try:
            out = subprocess.run([ctr, "oci", "spec"], capture_output=True,
                                 text=True, timeout=10)
            if out.returncode == 0:
                return json.loads(out.stdout)
        except (OSError, ValueError, subprocess.TimeoutExpired):
            pass
    return json.loads(json.dumps(DEFAULT_SPEC))  # deep copy


def merge_hook_into_spec(spec: dict, hook_path: str) -> bool:
    """Add the prestart hook; returns False when already present."""
    hooks = spec.setdefault("hooks", {})
    prestart = hooks.setdefault("prestart", [])
    if any(h.get("path") == hook_path for h in prestart):
        return False
    prestart.append({"path": hook_path, "args": [os.path.basename(hook_path),
                                                 "prestart"]})
    return True


def _parse_toml(text: str) -> dict:
    import tomli

    return tomli.loads(text)


def existing_base_spec(config: dict) -> str | None:
    """base_runtime_spec already configured on any runtime? Return its path."""
    plugins = config.get("plugins", {})
    for plug in plugins.values():
        runtimes = plug.get("containerd", {}).get("runtimes", {})
        for rt in runtimes.values():
            spec = rt.get("options", {}).get("base_runtime_spec")
            if spec:
                return spec
    return None


def patch_config_text(text: str, spec_path: str, runtime: str = "runc") -> str:
    """Insert base_runtime_spec into the runtime's options table (created
    after the runtime table if missing). Raises when the runtime table is
    absent entirely — that config isn't a CRI runtime config we understand."""
    line = f'  base_runtime_spec = "{spec_path}"\n'
    m = None
    for cand in OPTIONS_TABLE_RE.finditer(text):
        if cand.group("rt") == runtime:
            m = cand
            break
    if m is not None:
        insert_at = m.end()
        if insert_at < len(text) and text[insert_at] == "\n":
            insert_at += 1
        else:
            line = "\n" + line
        return text[:insert_at] + line + text[insert_at:]
    for cand in RUNTIME_TABLE_RE.finditer(text):
        if cand.group("rt") == runtime:
            cri = cand.group("cri")
            header = (f"[plugins.{cri}.containerd.runtimes.{runtime}.options]\n")
            end = text.find("\n", cand.end())
            end = len(text) if end < 0 else end + 1
            return text[:end] + header + line + text[end:]
    raise ValueError(f"no [...containerd.runtimes.{runtime}] table in config")


def install(config_path: str, spec_path: str, hook_path: str, runtime: str,
            dry_run: bool, host_root: str = "") -> int:
    # Paths INSIDE the TOML / spec are host-view (containerd resolves them on
    # the host); when running from the agent container, file access goes
    # through the /host mount instead — host_root bridges the two (the same
    # agent-view vs. advertise-view split as AgentPaths.limits_host_view).
    with open(config_path) as f:
        text = f.read()
    config = _parse_toml(text)

    target_spec = existing_base_spec(config)
    patch_toml = target_spec is None
    if target_spec is None:
        target_spec = spec_path
    spec_file = host_root + target_spec

    if os.path.exists(spec_file):
        with open(spec_file) as f:
            spec = json.load(f)
    else:
        spec = default_spec()
    spec_changed = merge_hook_into_spec(spec, hook_path)

    if patch_toml:
        new_text = patch_config_text(text, target_spec, runtime)
        _parse_toml(new_text)  # must stay valid TOML
        parsed = _parse_toml(new_text)
        if existing_base_spec(parsed) != target_spec:
            raise RuntimeError("patched config does not expose base_runtime_spec")
    else:
        new_text = text

    if dry_run:
        print(f"[dry-run] spec {'update' if spec_changed else 'unchanged'}: "
              f"{target_spec}")
        print(f"[dry-run] config {'patch' if patch_toml else 'unchanged'}: "
              f"{config_path}")
        return 0

    if spec_changed:
        tmp = spec_file + ".egpu-tmp"
        with open(tmp, "w") as f:
            json.dump(spec, f, indent=2)
        os.replace(tmp, spec_file)
        print(f"wrote {spec_file} (prestart hook: {hook_path})")
    else:
        print(f"{spec_file}: hook already registered")

    if patch_toml:
        shutil.copyfile(config_path, config_path + ".egpu-bak")
        tmp = config_path + ".egpu-tmp"
        with open(tmp, "w") as f:
            f.write(new_text)
        os.replace(tmp, config_path)
        print(f"patched {config_path} "
              f"(base_runtime_spec on runtime '{runtime}'; backup "
              f"{config_path}.egpu-bak)")
        print("restart containerd to take effect: systemctl restart containerd")
    elif spec_changed:
        print("existing base_runtime_spec reused; no config change, "
              "restart containerd to take effect")
    else:
        print("nothing to do")
    return 0


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    p.add_argument("--config", default="/etc/containerd/config.toml")
    p.add_argument("--spec", default="/etc/containerd/egpu-base.json",
                   help="base spec to create when none is configured "
                        "(HOST-view path — what goes into the TOML)")
    p.add_argument("--hook", default=HOOK_DEFAULT)
    p.add_argument("--runtime", default="runc")
    p.add_argument("--host-root", default="",
                   help="prefix for FILE access to host-view paths when "
                        "running inside the agent container (e.g. /host)")
    p.add_argument("--dry-run", action="store_true")
    args = p.parse_args(argv)
    try:
        return install(args.config, args.spec, args.hook, args.runtime,
                       args.dry_run, host_root=args.host_root)
    except (OSError, ValueError, RuntimeError) as e:
        print(f"error: {e}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
