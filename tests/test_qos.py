"""QoS-class plumbing: pod annotation → limits file → (shim applies queue
priority in-container; GPU-side application is covered by the shim itself)."""
from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.kube.pods import Pod
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import Harness


def test_qos_annotation_parsing():
    p = Pod(namespace="ns", name="p", annotations={consts.ELASTIC_GPU_QOS_ANNOTATION: "high"})
    assert p.qos_class() == "high"
    p2 = Pod(namespace="ns", name="p", annotations={consts.ELASTIC_GPU_QOS_ANNOTATION: "bogus"})
    assert p2.qos_class() is None
    assert Pod(namespace="ns", name="p").qos_class() is None


def test_priority_lands_in_limits(tmp_path):
    h = Harness(str(tmp_path), gpus=1)
    ids = [f"0-{i:02d}" for i in range(20)]
    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
    h.core_locator.assign(d.hash, PodContainer("ns", "pq", "main"))
    pod = h.add_assumed_pod("ns", "pq", "main", "0")
    pod.annotations[consts.ELASTIC_GPU_QOS_ANNOTATION] = "low"
    h.plugin.core.allocate({"container_requests": [{"devicesIDs": ids}]}, None)
    h.plugin.core.pre_start_container({"devicesIDs": ids}, None)
    limits = h.plugin.cfg.limits.read(d.hash)
    assert limits["priority"] == "low"
    h.close()


def test_no_priority_key_without_annotation(tmp_path):
    h = Harness(str(tmp_path), gpus=1)
    ids = [f"0-{i:02d}" for i in range(20)]
    d = Device.new(ids, consts.RESOURCE_GPU_CORE)
    h.core_locator.assign(d.hash, PodContainer("ns", "pn", "main"))
    h.add_assumed_pod("ns", "pn", "main", "0")
    h.plugin.core.allocate({"container_requests": [{"devicesIDs": ids}]}, None)
    h.plugin.core.pre_start_container({"devicesIDs": ids}, None)
    assert "priority" not in h.plugin.cfg.limits.read(d.hash)
    h.close()
