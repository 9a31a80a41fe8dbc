"""Scheduler-sim tests: placement through the agent's live
GetPreferredAllocation RPC, annotation contract, full bind loop."""
import pytest

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.operator.fake import FakeBackend
from elastic_gpu_agent_amd.schedsim import SimScheduler
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import Harness, PluginClient


def test_place_fractional_pods_pack_one_gpu():
    sched = SimScheduler(FakeBackend(count=2).devices())
    a = sched.place("main", core_units=30)
    b = sched.place("main", core_units=30)
    assert a and b
    # both fractions land on the same GPU (packing), leaving GPU 1 whole
    assert a["gpu_indexes"] == b["gpu_indexes"]
    assert a["annotations"][consts.ELASTIC_GPU_ASSUMED_ANNOTATION] == "true"
    assert set(a["core_ids"]).isdisjoint(b["core_ids"])


def test_place_whole_and_reject_overflow():
    sched = SimScheduler(FakeBackend(count=1).devices())
    whole = sched.place("main", core_units=100)
    assert whole and len(whole["core_ids"]) == 100
    assert sched.place("main", core_units=1) is None  # node full
    sched.release(core_ids=whole["core_ids"])
    assert sched.place("main", core_units=1) is not None


def test_place_multi_gpu():
    sched = SimScheduler(FakeBackend(count=2).devices())
    r = sched.place("main", core_units=200)
    assert r and len(r["gpu_indexes"]) == 2
    assert r["annotations"][consts.ELASTIC_GPU_CONTAINER_ANNOTATION % "main"] in (
        "0,1", "1,0")


def test_schedsim_through_live_agent(tmp_path):
    """Placement via the agent's GetPreferredAllocation RPC, then the full
    bind loop with the produced annotations."""
    h = Harness(str(tmp_path), gpus=2)
    try:
        h.plugin.core_server.serve()
        h.plugin.core_server.wait_ready()
        client = PluginClient(h.plugin.core_server.socket_path)

        def preferred(resource, avail, size):
            resp = client.preferred({
                "container_requests": [{
                    "available_deviceIDs": avail,
                    "must_include_deviceIDs": [],
                    "allocation_size": size,
                }]
            })
            return resp["container_responses"][0]["deviceIDs"]

        sched = SimScheduler(h.operator.devices())
        placement = sched.place("main", core_units=25, preferred_fn=preferred)
        assert placement is not None

        ids = placement["core_ids"]
        d = Device.new(ids, consts.RESOURCE_GPU_CORE)
        h.core_locator.assign(d.hash, PodContainer("ns", "sched-pod", "main"))
        from elastic_gpu_agent_amd.kube.pods import Pod

        h.sitter.add(Pod(namespace="ns", name="sched-pod",
                         annotations=placement["annotations"]))
        client.allocate({"container_requests": [{"devicesIDs": ids}]})
        client.pre_start({"devicesIDs": ids})
        pi = h.storage.load("ns", "sched-pod")
        assert pi.container_device_map["main"].hash == d.hash
        client.close()
    finally:
        h.close()
