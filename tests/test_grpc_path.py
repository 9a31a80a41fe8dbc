"""Full gRPC-path tests over unix sockets: serve/register loop, ListAndWatch
stream, Allocate/PreStart RPCs, podresources locator, kubelet-restart
re-registration."""
import os
import time

import grpc
import pytest

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.kube.locator import KubeletDeviceLocator
from elastic_gpu_agent_amd.kube.podresources_server import PodResourcesServer
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import FakeKubeletRegistration, Harness, PluginClient


@pytest.fixture
def h(tmp_path):
    harness = Harness(str(tmp_path), gpus=2)
    yield harness
    harness.close()


def test_serve_register_and_rpcs(h, tmp_path):
    kubelet = FakeKubeletRegistration(h.paths.kubelet_socket)
    kubelet.start()
    try:
        h.plugin.run()
        assert h.plugin.core_server.wait_registered(10)
        assert h.plugin.memory_server.wait_registered(10)
        assert kubelet.wait_for_register(2)
        resources = {r["resource_name"] for r in kubelet.requests}
        assert resources == {consts.RESOURCE_GPU_CORE, consts.RESOURCE_GPU_MEMORY}
        req = next(r for r in kubelet.requests if r["resource_name"] == consts.RESOURCE_GPU_CORE)
        assert req["version"] == "v1beta1"
        assert req["endpoint"] == consts.CORE_SOCK_NAME
        assert req["options"]["pre_start_required"] is True

        client = PluginClient(h.plugin.core_server.socket_path)
        try:
            # options RPC
            opts = client.get_options({})
            assert opts["get_preferred_allocation_available"] is True

            # ListAndWatch first message
            stream = client.list_and_watch({})
            first = next(stream)
            assert len(first["devices"]) == 200
            stream.close()  # generator: releases the channel for unary calls

            # Allocate + PreStart over the wire
            ids = [f"1-{i:02d}" for i in range(40)]
            d = Device.new(ids, consts.RESOURCE_GPU_CORE)
            h.core_locator.assign(d.hash, PodContainer("ns", "gp", "main"))
            h.add_assumed_pod("ns", "gp", "main", "1")
            resp = client.allocate({"container_requests": [{"devicesIDs": ids}]})
            assert resp["container_responses"][0]["envs"]["GPU"] == d.hash
            client.pre_start({"devicesIDs": ids})
            assert os.readlink(
                os.path.join(h.paths.dev_root, f"elastic-gpu-{d.hash}-0")
            ) == "/dev/dri/renderD129"

            # error path surfaces as INVALID_ARGUMENT
            from elastic_gpu_agent_amd import egrpc

            bad = [f"0-{i:02d}" for i in range(10)]
            with pytest.raises(egrpc.EgrpcError) as ei:
                client.pre_start({"devicesIDs": bad})
            assert ei.value.code() == egrpc.INVALID_ARGUMENT
        finally:
            client.close()

        # ---- the same socket driven by the REAL gRPC stack (kubelet dir.) --
        from helpers import GrpcioPluginClient

        gclient = GrpcioPluginClient(h.plugin.core_server.socket_path)
        try:
            opts = gclient.get_options({})
            assert opts["pre_start_required"] is True
            stream = gclient.list_and_watch({})
            first = next(stream)
            assert len(first["devices"]) == 200
            stream.cancel()
            ids2 = [f"0-{i:02d}" for i in range(60, 80)]
            d2 = Device.new(ids2, consts.RESOURCE_GPU_CORE)
            h.core_locator.assign(d2.hash, PodContainer("ns", "gp2", "main"))
            h.add_assumed_pod("ns", "gp2", "main", "0")
            resp = gclient.allocate({"container_requests": [{"devicesIDs": ids2}]})
            assert resp["container_responses"][0]["envs"]["GPU"] == d2.hash
            gclient.pre_start({"devicesIDs": ids2})
            with pytest.raises(grpc.RpcError) as gei:
                gclient.pre_start({"devicesIDs": ["9-99"]})
            assert gei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
        finally:
            gclient.close()
    finally:
        kubelet.stop()


def test_reregister_after_kubelet_restart(h):
    kubelet = FakeKubeletRegistration(h.paths.kubelet_socket)
    kubelet.start()
    try:
        h.plugin.core_server.start()
        assert h.plugin.core_server.wait_registered(10)
        assert kubelet.wait_for_register(1)
        n0 = len(kubelet.requests)

        # kubelet restart: old socket replaced by a new one (new inode).
        # A real kubelet restart takes seconds; recreate after >1 watcher
        # tick so the (inode, ctime) identity check can't alias a same-tick
        # recreation with a recycled inode (tmpfs does recycle them).
        kubelet.stop()
        if os.path.exists(h.paths.kubelet_socket):  # grpc may remove it on stop
            os.unlink(h.paths.kubelet_socket)
        time.sleep(1.5)
        kubelet2 = FakeKubeletRegistration(h.paths.kubelet_socket)
        kubelet2.start()
        try:
            assert kubelet2.wait_for_register(1, timeout=40.0)
            assert len(kubelet2.requests) >= 1
        finally:
            kubelet2.stop()
    finally:
        kubelet.stop()


def test_locator_against_podresources_server(tmp_path):
    sock = str(tmp_path / "podresources.sock")
    server = PodResourcesServer(sock)
    server.start()
    try:
        ids = [f"0-{i:02d}" for i in range(30)]
        # >=1.21 kubelet shape: one ID per ContainerDevices entry
        for did in ids:
            server.set_assignment("ns", "pod-a", "main", consts.RESOURCE_GPU_CORE, [did])
        # unrelated pod with another resource
        server.set_assignment("ns", "pod-b", "main", "other/resource", ["0-00"])

        loc = KubeletDeviceLocator(consts.RESOURCE_GPU_CORE, sock)
        pc = loc.locate(Device.new(ids, consts.RESOURCE_GPU_CORE))
        assert (pc.namespace, pc.name, pc.container) == ("ns", "pod-a", "main")

        # <=1.20 shape: all IDs in one entry
        server.remove_pod("ns", "pod-a")
        server.set_assignment("ns", "pod-c", "c1", consts.RESOURCE_GPU_CORE, ids)
        pc = loc.locate(Device.new(ids, consts.RESOURCE_GPU_CORE))
        assert pc.name == "pod-c"

        with pytest.raises(KeyError):
            loc.locate(Device.new(["9-99"], consts.RESOURCE_GPU_CORE))
        loc.close()
    finally:
        server.stop()


def test_locator_lazy_reconnect(tmp_path):
    """Locator survives a podresources server restart (lazy re-dial)."""
    sock = str(tmp_path / "podresources.sock")
    server = PodResourcesServer(sock)
    server.start()
    loc = KubeletDeviceLocator(consts.RESOURCE_GPU_CORE, sock)
    ids = ["0-00"]
    server.set_assignment("ns", "p", "c", consts.RESOURCE_GPU_CORE, ids)
    assert loc.locate(Device.new(ids, consts.RESOURCE_GPU_CORE)).name == "p"

    server.stop()
    if os.path.exists(sock):
        os.unlink(sock)
    server2 = PodResourcesServer(sock)
    server2.set_assignment("ns", "p2", "c", consts.RESOURCE_GPU_CORE, ids)
    server2.start()
    try:
        assert loc.locate(Device.new(ids, consts.RESOURCE_GPU_CORE)).name == "p2"
    finally:
        loc.close()
        server2.stop()


def test_preferred_allocation_over_the_wire(h):
    """GetPreferredAllocation through the served socket — exercising the
    digest deserializer + raw-body serializer exactly as kubelet would —
    from BOTH client stacks (egrpc and grpcio)."""
    from helpers import GrpcioPluginClient

    h.plugin.core_server.serve()
    h.plugin.memory_server.serve()
    try:
        for mk in (PluginClient, GrpcioPluginClient):
            core = mk(h.plugin.core_server.socket_path)
            mem = mk(h.plugin.memory_server.socket_path)
            try:
                avail = [f"{g}-{i:02d}" for g in range(2) for i in range(100)]
                resp = core.preferred({"container_requests": [
                    {"available_deviceIDs": avail, "allocation_size": 30}]})
                ids = resp["container_responses"][0]["deviceIDs"]
                assert len(ids) == 30
                assert len({i.split("-")[0] for i in ids}) == 1  # single GPU

                # multi-GPU core request (falls back to the generic policy)
                resp = core.preferred({"container_requests": [
                    {"available_deviceIDs": avail, "allocation_size": 200}]})
                assert len(resp["container_responses"][0]["deviceIDs"]) == 200

                # memory request with a larger pool
                mem_avail = [f"{g}-{i:06d}" for g in range(2) for i in range(5000)]
                resp = mem.preferred({"container_requests": [
                    {"available_deviceIDs": mem_avail, "allocation_size": 4096},
                    {"available_deviceIDs": [], "allocation_size": 1},
                ]})
                ids = resp["container_responses"][0]["deviceIDs"]
                assert len(ids) == 4096
                assert ids == sorted(ids)
                assert len({i.split("-")[0] for i in ids}) == 1
                assert resp["container_responses"][1].get("deviceIDs", []) == []

                # must_include honored through the wire (generic fallback)
                resp = core.preferred({"container_requests": [
                    {"available_deviceIDs": avail,
                     "must_include_deviceIDs": ["1-07"],
                     "allocation_size": 10}]})
                ids = resp["container_responses"][0]["deviceIDs"]
                assert "1-07" in ids and len(ids) == 10
                assert all(i.startswith("1-") for i in ids)
            finally:
                core.close()
                mem.close()
    finally:
        h.plugin.core_server.stop()
        h.plugin.memory_server.stop()
