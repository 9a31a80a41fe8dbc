"""Node-level end-to-end test (BASELINE config #1 equivalent, in-process):

real GPUManager wiring — PodSitter watching a stub k8s API server, real
KubeletDeviceLocator against an in-process podresources server, both device
plugins served on real unix sockets and registered with a fake kubelet —
driving a pod through its entire life: schedule (annotations) → Allocate →
PreStart (symlinks+mask+limits) → OCI hook dry-run → delete → event-driven GC.
"""
import json
import os
import subprocess
import time

import pytest

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.kube.client import K8sClient
from elastic_gpu_agent_amd.kube.locator import KubeletDeviceLocator
from elastic_gpu_agent_amd.kube.podresources_server import PodResourcesServer
from elastic_gpu_agent_amd.manager import GPUManager, ManagerOptions
from elastic_gpu_agent_amd.plugins.config import AgentPaths, PluginOptions
from elastic_gpu_agent_amd.types import Device

from helpers import FakeKubeletRegistration, PluginClient
from test_kube_client import StubK8s

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HOOK = os.path.join(REPO, "bin", "egpu-hook")


@pytest.fixture
def node(tmp_path):
    """A fully wired single-node environment."""
    stub = StubK8s()
    plugin_dir = str(tmp_path / "device-plugins")
    os.makedirs(plugin_dir)
    podres_sock = str(tmp_path / "podresources.sock")
    podres = PodResourcesServer(podres_sock)
    podres.start()
    kubelet = FakeKubeletRegistration(os.path.join(plugin_dir, "kubelet.sock"))
    kubelet.start()

    paths = AgentPaths(
        dev_root=str(tmp_path / "dev"),
        plugin_dir=plugin_dir,
        podresources_socket=podres_sock,
        limits_dir=str(tmp_path / "limits"),
        state_dir=str(tmp_path / "state"),
        shim_host_path=str(tmp_path / "libegpu_shim.so"),
    )
    client = K8sClient(base_url=f"http://127.0.0.1:{stub.port}")
    from elastic_gpu_agent_amd.kube.sitter import PodSitter

    mgr = GPUManager.__new__(GPUManager)  # wire manually with our client
    opts = ManagerOptions(
        node_name="n1",
        db_path=str(tmp_path / "meta.db"),
        backend="fake",
        paths=paths,
        plugin_options=PluginOptions(mem_unit_mib=1024),
    )
    locators = (
        KubeletDeviceLocator(consts.RESOURCE_GPU_CORE, podres_sock),
        KubeletDeviceLocator(consts.RESOURCE_GPU_MEMORY, podres_sock),
    )
    sitter = PodSitter(client, "n1", delete_hook=None)
    GPUManager.__init__(mgr, opts, sitter=None or sitter, locators=locators)
    # re-hook delete events into the manager's gc queue
    sitter._hook = mgr._on_pod_delete

    yield {
        "stub": stub, "podres": podres, "kubelet": kubelet, "mgr": mgr,
        "paths": paths, "tmp": tmp_path,
    }
    mgr.stop()
    kubelet.stop()
    podres.stop()
    stub.stop()


@pytest.mark.timeout(120)
def test_full_pod_lifecycle(node):
    stub, podres, kubelet, mgr = (node["stub"], node["podres"], node["kubelet"],
                                  node["mgr"])
    paths = node["paths"]
    mgr.run()
    assert mgr.plugin.core_server.wait_registered(15)
    assert mgr.plugin.memory_server.wait_registered(15)
    assert kubelet.wait_for_register(2)

    # kubelet view: both resources advertised (8 fake GPUs)
    client = PluginClient(mgr.plugin.core_server.socket_path)
    stream = client.list_and_watch({})
    first = next(stream)
    assert len(first["devices"]) == 800
    stream.close()

    # scheduler assumes the pod onto GPU 2 with 40% core
    ids = [f"2-{i:02d}" for i in range(40)]
    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
    pod = stub.add_pod(
        "default", "workload", node="n1",
        annotations={
            consts.ELASTIC_GPU_ASSUMED_ANNOTATION: "true",
            consts.ELASTIC_GPU_CONTAINER_ANNOTATION % "main": "2",
        },
    )
    stub.push_event("ADDED", pod)

    # kubelet allocates + records podresources + prestarts
    resp = client.allocate({"container_requests": [{"devicesIDs": ids}]})
    cr = resp["container_responses"][0]
    assert cr["envs"]["GPU"] == d.hash
    assert cr["envs"]["HSA_TOOLS_LIB"] == paths.shim_container_path
    podres.set_assignment("default", "workload", "main", consts.RESOURCE_GPU_CORE, ids)

    deadline = time.time() + 10  # wait for the sitter to see the pod
    while time.time() < deadline:
        try:
            mgr.sitter.get_pod("default", "workload")
            break
        except Exception:
            time.sleep(0.05)
    client.pre_start({"devicesIDs": ids})

    gpu_link = os.path.join(paths.dev_root, f"elastic-gpu-{d.hash}-0")
    assert os.readlink(gpu_link) == "/dev/dri/renderD130"  # fake GPU 2
    limits = json.loads(open(mgr.config.limits.host_path(d.hash)).read())
    assert limits["cu_count"] == 102  # 40% of 256 → 102.4 → 102 (51 CU pairs)

    # OCI hook (dry-run) resolves the same allocation into device nodes
    bundle = node["tmp"] / "bundle"
    bundle.mkdir()
    (bundle / "config.json").write_text(json.dumps(
        {"process": {"env": [f"GPU={d.hash}"]}}))
    env = dict(os.environ)
    env.update({
        "EGPU_HOOK_DRYRUN": "1",
        "EGPU_DEV_ROOT": paths.dev_root,
        "EGPU_HOOK_LOG": str(node["tmp"] / "hook.log"),
        "EGPU_STATE_DIR": paths.state_dir,
    })
    r = subprocess.run(
        [HOOK, "prestart"],
        input=json.dumps({"pid": os.getpid(), "bundle": str(bundle)}).encode(),
        env=env, capture_output=True, timeout=30,
    )
    assert r.returncode == 0, r.stderr
    plan = json.loads(r.stdout)
    assert any(n["minor"] == 130 for n in plan["nodes"])
    # hook recorded the pid for occupancy attribution
    assert (node["tmp"] / "state" / "pids" / d.hash).exists()

    # state persisted
    pi = mgr.storage.load("default", "workload")
    assert pi.container_device_map["main"].hash == d.hash

    # pod deleted → watch event → event-driven GC reclaims everything
    stub.pods.pop(("default", "workload"))
    stub.push_event("DELETED", pod)
    deadline = time.time() + 20
    while os.path.lexists(gpu_link) and time.time() < deadline:
        time.sleep(0.1)
    assert not os.path.lexists(gpu_link), "event-driven GC did not reclaim"
    with pytest.raises(KeyError):
        mgr.storage.load("default", "workload")
    assert not (node["tmp"] / "state" / "pids" / d.hash).exists()
    client.close()
