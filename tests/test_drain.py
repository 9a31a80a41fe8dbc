"""Drain / repartition workflow (elastic_gpu_agent_amd/drain.py).

The reference never re-checks device health after the first ListAndWatch
send, so it has no way to take a GPU out of scheduling; the
drain flag + per-GPU Unhealthy advertisement is the MI355X-side workflow
that makes compute-partition changes (SPX↔CPX) operable.
"""
from __future__ import annotations

import json

import pytest

from elastic_gpu_agent_amd import consts
from elastic_gpu_agent_amd.cli import egpuctl
from elastic_gpu_agent_amd.drain import (
    clear_drain,
    drained_indexes,
    list_drains,
    live_allocations_on,
    set_drain,
    wait_drained,
)
from elastic_gpu_agent_amd.protos import deviceplugin as dp
from elastic_gpu_agent_amd.types import Device, PodContainer

from helpers import Harness


def _health_by_gpu(devices):
    """{gpu_index: {healths seen}} from a ListAndWatch device list."""
    out = {}
    for d in devices:
        idx = int(d["ID"].split("-")[0])
        out.setdefault(idx, set()).add(d["health"])
    return out


def test_drained_gpu_advertised_unhealthy(tmp_path):
    h = Harness(str(tmp_path), gpus=2)
    try:
        set_drain(h.storage, 0)
        for plugin in (h.plugin.core, h.plugin.memory):
            by_gpu = _health_by_gpu(plugin.list_devices(True))
            assert by_gpu[0] == {consts.UNHEALTHY}
            assert by_gpu[1] == {consts.HEALTHY}
        clear_drain(h.storage, 0)
        by_gpu = _health_by_gpu(h.plugin.core.list_devices(True))
        assert by_gpu[0] == {consts.HEALTHY}
    finally:
        h.close()


def test_drained_gpu_in_encoded_snapshot(tmp_path):
    """The fastpath (pre-encoded) snapshot must agree with the dict path."""
    h = Harness(str(tmp_path), gpus=2)
    try:
        set_drain(h.storage, 1)
        encoded = h.plugin.core.list_and_watch_encoded(None)
        snap = dp.ListAndWatchResponse.decode(next(encoded))
        by_gpu = _health_by_gpu(snap["devices"])
        assert by_gpu[1] == {consts.UNHEALTHY}
        assert by_gpu[0] == {consts.HEALTHY}
    finally:
        h.close()


def test_drain_flag_changes_snapshot_on_refresh(tmp_path):
    """_watch_snapshots re-advertises when the drain flag flips: the fresh
    snapshot differs from the cached one, which is exactly the re-advertise
    condition of the watch loop."""
    h = Harness(str(tmp_path), gpus=2)
    try:
        before = h.plugin.core.list_devices(True)
        set_drain(h.storage, 0)
        after = h.plugin.core.list_devices(True)
        assert before != after
        clear_drain(h.storage, 0)
        assert h.plugin.core.list_devices(True) == before
    finally:
        h.close()


def _bind_fractional(h: Harness, ns, name, container, gpu_index, percent):
    ids = [f"{gpu_index}-{s:02d}" for s in range(percent)]
    device = Device.new(ids, consts.RESOURCE_GPU_CORE)
    h.core_locator.assign(device.hash, PodContainer(ns, name, container))
    h.add_assumed_pod(ns, name, container, str(gpu_index))
    h.plugin.core.allocate({"container_requests": [{"devicesIDs": ids}]}, None)
    h.plugin.core.pre_start_container({"devicesIDs": ids}, None)
    return device


def test_live_allocations_join(tmp_path):
    h = Harness(str(tmp_path), gpus=2)
    try:
        _bind_fractional(h, "ns", "p0", "c", 0, 25)
        _bind_fractional(h, "ns", "p1", "c", 1, 50)
        limits = h.plugin.core.cfg.limits
        on0 = live_allocations_on(h.storage, 0, limits)
        on1 = live_allocations_on(h.storage, 1, limits)
        assert [r["pod"] for r in on0] == ["ns/p0"]
        assert [r["pod"] for r in on1] == ["ns/p1"]
        assert live_allocations_on(h.storage, 7, limits) == []
    finally:
        h.close()


def test_live_allocations_whole_gpu_via_limits(tmp_path):
    """Whole-GPU core allocations have no CU-mask record; attribution comes
    from the limits file's gpu_indexes."""
    h = Harness(str(tmp_path), gpus=2)
    try:
        _bind_fractional(h, "ns", "whole", "c", 1, 100)
        limits = h.plugin.core.cfg.limits
        assert [r["pod"] for r in live_allocations_on(h.storage, 1, limits)] == [
            "ns/whole"
        ]
        assert live_allocations_on(h.storage, 0, limits) == []
    finally:
        h.close()


def test_wait_drained(tmp_path):
    h = Harness(str(tmp_path), gpus=2)
    try:
        _bind_fractional(h, "ns", "p0", "c", 0, 25)
        limits = h.plugin.core.cfg.limits
        # still occupied -> times out quickly, flag semantics unchanged
        assert not wait_drained(h.storage, 0, limits, timeout=0.05, poll_interval=0.01)
        # pod goes away + GC reclaims -> drained
        h.sitter.remove("ns", "p0")
        assert h.plugin.gc_once() == 1
        assert wait_drained(h.storage, 0, limits, timeout=1.0, poll_interval=0.01)
    finally:
        h.close()


def test_drain_cli_roundtrip(tmp_path, capsys):
    h = Harness(str(tmp_path), gpus=2)
    try:
        db = h.storage.path
        limits_dir = str(tmp_path / "limits")
        rc = egpuctl.main(["--db", db, "--limits-dir", limits_dir, "drain", "0"])
        assert rc == 0
        assert drained_indexes(h.storage) == {0}
        rc = egpuctl.main(["--db", db, "undrain", "0"])
        assert rc == 0
        assert drained_indexes(h.storage) == set()
    finally:
        h.close()


def test_drain_cli_list_json(tmp_path, capsys):
    h = Harness(str(tmp_path), gpus=2)
    try:
        set_drain(h.storage, 1, mode="CPX")
        rc = egpuctl.main(["--db", h.storage.path, "drain", "--list"])
        assert rc == 0
        out = json.loads(capsys.readouterr().out)
        assert out["1"]["requested_mode"] == "CPX"
        assert list_drains(h.storage)[1]["requested_mode"] == "CPX"
    finally:
        h.close()


def test_drain_cli_wait_timeout(tmp_path):
    h = Harness(str(tmp_path), gpus=2)
    try:
        _bind_fractional(h, "ns", "p0", "c", 0, 25)
        rc = egpuctl.main(
            ["--db", h.storage.path, "--limits-dir",
             str(tmp_path / "limits"), "drain", "0", "--wait", "--timeout", "0.05"]
        )
        assert rc == 1  # occupied: timeout, flag stays set
        assert drained_indexes(h.storage) == {0}
    finally:
        h.close()


def test_drain_stops_simulated_scheduling(tmp_path):
    """End-to-end with the scheduler sim: after a drain + health sync (what
    kubelet does with the Unhealthy re-advertisement), new pods land on the
    other GPU only; undrain + sync restores placement."""
    from elastic_gpu_agent_amd.schedsim import SimScheduler

    h = Harness(str(tmp_path), gpus=2)
    try:
        sim = SimScheduler(devices=h.operator.devices(), mem_unit_mib=1024)
        set_drain(h.storage, 0)
        sim.sync_health(h.plugin.core, h.plugin.memory)
        for i in range(3):
            placed = sim.place(f"c{i}", core_units=30)
            assert placed is not None
            assert placed["gpu_indexes"] == [1], placed
        # GPU 1 has 10 core units left; a 30-unit pod no longer fits anywhere
        assert sim.place("cx", core_units=30) is None
        clear_drain(h.storage, 0)
        sim.sync_health(h.plugin.core, h.plugin.memory)
        placed = sim.place("cy", core_units=30)
        assert placed is not None and placed["gpu_indexes"] == [0]
    finally:
        h.close()


def test_drain_wait_repartition_cycle(tmp_path, monkeypatch):
    """Full lifecycle: drain → GPU empties → partition flipped via amdsmi →
    flag cleared (GPU back in service)."""
    import elastic_gpu_agent_amd as pkg

    calls = []

    class StubSmi:
        @staticmethod
        def set_compute_partition(index, mode):
            calls.append((index, mode))

        @staticmethod
        def get_compute_partition(index):
            return calls[-1][1] if calls else "SPX"

    monkeypatch.setattr(pkg, "_amdsmi", StubSmi, raising=False)
    h = Harness(str(tmp_path), gpus=2)
    try:
        _bind_fractional(h, "ns", "p0", "c", 0, 25)

        import threading

        def finish_pod():
            import time

            time.sleep(0.15)
            h.sitter.remove("ns", "p0")
            h.plugin.gc_once()

        t = threading.Thread(target=finish_pod)
        t.start()
        rc = egpuctl.main(
            ["--db", h.storage.path, "--limits-dir", str(tmp_path / "limits"),
             "drain", "0", "--wait", "--timeout", "5", "--repartition", "CPX"]
        )
        t.join()
        assert rc == 0
        assert calls == [(0, "CPX")]
        assert drained_indexes(h.storage) == set()  # undrained after flip
    finally:
        h.close()


def test_drain_survives_restart(tmp_path):
    """The drain flag lives in the persisted aux table: a restarted agent
    (fresh plugin over the same DB) still advertises the GPU Unhealthy."""
    h = Harness(str(tmp_path), gpus=2)
    try:
        set_drain(h.storage, 0)
    finally:
        h.close()
    h2 = Harness(str(tmp_path), gpus=2)
    try:
        by_gpu = _health_by_gpu(h2.plugin.core.list_devices(True))
        assert by_gpu[0] == {consts.UNHEALTHY}
        assert by_gpu[1] == {consts.HEALTHY}
    finally:
        h2.close()


def test_drain_cli_requires_index(tmp_path):
    h = Harness(str(tmp_path), gpus=1)
    try:
        assert egpuctl.main(["--db", h.storage.path, "drain"]) == 2
    finally:
        h.close()
