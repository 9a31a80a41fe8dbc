"""k8s Event emission on binding failures."""
import pytest

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import Harness


def test_bind_failure_emits_event(tmp_path):
    h = Harness(str(tmp_path), gpus=1)
    events = []
    h.plugin.cfg.event_sink = lambda ns, pod, reason, msg: events.append(
        (ns, pod, reason, msg)
    )
    # 150-unit allocation (needs 2 GPUs) but annotation names only one index
    ids = [f"0-{i:02d}" for i in range(100)] + [f"0-{i:02d}" for i in range(50)]
    ids = [f"0-{i:02d}" for i in range(100)]
    ids += [f"1-{i:02d}" for i in range(50)]  # but harness has 1 GPU; bind fails
    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
    h.core_locator.assign(d.hash, PodContainer("ns", "bad", "main"))
    h.add_assumed_pod("ns", "bad", "main", "0")  # 1 index, 2 links needed
    with pytest.raises(RuntimeError):
        h.plugin.core.pre_start_container({"devicesIDs": ids}, None)
    assert len(events) == 1
    ns, pod, reason, msg = events[0]
    assert (ns, pod, reason) == ("ns", "bad", "EgpuBindFailed")
    assert d.hash in msg
    h.close()


def test_event_sink_failure_does_not_break_success_path(tmp_path):
    h = Harness(str(tmp_path), gpus=1)

    def broken_sink(*a):
        raise RuntimeError("events down")

    h.plugin.cfg.event_sink = broken_sink
    ids = [f"0-{i:02d}" for i in range(10)]
    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
    h.core_locator.assign(d.hash, PodContainer("ns", "ok", "main"))
    h.add_assumed_pod("ns", "ok", "main", "0")
    h.plugin.core.allocate({"container_requests": [{"devicesIDs": ids}]}, None)
    h.plugin.core.pre_start_container({"devicesIDs": ids}, None)  # no raise
    h.close()


def test_k8s_client_create_event(tmp_path):
    """Against the stub API server (records POSTs)."""
    import sys, os

    sys.path.insert(0, os.path.dirname(__file__))
    from test_kube_client import StubK8s

    # extend stub with POST handling
    stub = StubK8s()
    posted = []
    orig_handler = stub.server.RequestHandlerClass

    class H(orig_handler):
        def do_POST(self):
            import json as _json

            length = int(self.headers.get("Content-Length", 0))
            posted.append(_json.loads(self.rfile.read(length)))
            body = b"{}"
            self.send_response(201)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    stub.server.RequestHandlerClass = H
    try:
        from elastic_gpu_agent_amd.kube.client import K8sClient

        c = K8sClient(base_url=f"http://127.0.0.1:{stub.port}")
        c.create_event("ns", "pod-x", "uid-1", "EgpuBindFailed", "boom")
        assert posted and posted[0]["involvedObject"]["name"] == "pod-x"
        assert posted[0]["reason"] == "EgpuBindFailed"
        assert posted[0]["source"]["component"] == "elastic-gpu-agent-amd"
        c.close()
    finally:
        stub.stop()
