"""GPU drain / repartition workflow (agent-side).

Closes the CPX-lifecycle gap (ROADMAP item 2): switching a MI355X between
compute partitions (SPX ↔ CPX etc.) requires the GPU to be idle, so the
operator needs a way to stop NEW placements while existing pods finish.

Mechanism — no new control channel, just the shared state DB:

- ``egpuctl drain <idx>`` writes an aux flag ``drain/<idx>``.
- Both device-plugin servers read the flag when building a ListAndWatch
  snapshot and advertise every fake device of a drained GPU **Unhealthy**;
  kubelet then stops placing pods there (running pods are untouched — device
  health only gates scheduling). The agent's periodic enumeration refresh
  (`GPUSharePluginBase._watch_snapshots`) notices the changed snapshot and
  re-advertises within ``--health-refresh-seconds``, so the CLI and the
  daemon need no direct connection.
- ``egpuctl drain <idx> --wait --repartition CPX`` polls the allocation
  state until no live allocation references the GPU, flips the compute
  partition via amdsmi, then clears the flag — the agent re-advertises the
  new geometry on its next refresh.

The reference has no equivalent (its device list is advertised once and
never re-checked — SURVEY §3.2); this is MI355X-scoped, built on the same
storage layer that already survives agent restarts.
"""
from __future__ import annotations

import json
import time
from typing import List, Optional, Set

from .isolation import AUX_MASK_PREFIX

AUX_DRAIN_PREFIX = "drain/"


def set_drain(storage, gpu_index: int, mode: Optional[str] = None) -> None:
    storage.aux_set(
        AUX_DRAIN_PREFIX + str(gpu_index),
        json.dumps({"since": int(time.time()), "requested_mode": mode}),
    )


def clear_drain(storage, gpu_index: int) -> None:
    storage.aux_delete(AUX_DRAIN_PREFIX + str(gpu_index))


def drained_indexes(storage) -> Set[int]:
    out: Set[int] = set()
    for key, _ in storage.aux_items(AUX_DRAIN_PREFIX):
        try:
            out.add(int(key[len(AUX_DRAIN_PREFIX):]))
        except ValueError:
            continue
    return out


def list_drains(storage) -> dict:
    out = {}
    for key, val in storage.aux_items(AUX_DRAIN_PREFIX):
        try:
            out[int(key[len(AUX_DRAIN_PREFIX):])] = json.loads(val)
        except ValueError:
            continue
    return out


def live_allocations_on(storage, gpu_index: int, limits_writer=None) -> List[dict]:
    """Allocations still bound to ``gpu_index``: storage records joined with
    the CU-mask aux table (fractional core) and the limits files (whole-GPU /
    memory allocations, which carry ``gpu_indexes`` but no mask)."""
    masks = {
        k[len(AUX_MASK_PREFIX):]: json.loads(v)
        for k, v in storage.aux_items(AUX_MASK_PREFIX)
    }
    rows: List[dict] = []

    def visit(pi):
        for container, device in pi.container_device_map.items():
            indexes: List[int] = []
            rec = masks.get(device.hash)
            if rec is not None and rec.get("gpu_index") is not None:
                indexes = [rec["gpu_index"]]
            elif limits_writer is not None:
                try:
                    indexes = limits_writer.read(device.hash).get("gpu_indexes") or []
                except (OSError, ValueError):
                    indexes = []
            if gpu_index in indexes:
                rows.append(
                    {
                        "pod": pi.key(),
                        "container": container,
                        "hash": device.hash,
                        "resource": device.resource_name,
                    }
                )

    storage.for_each(visit)
    return rows


def wait_drained(
    storage,
    gpu_index: int,
    limits_writer=None,
    timeout: float = 600.0,
    poll_interval: float = 2.0,
    progress=None,
) -> bool:
    """Block until no live allocation references the GPU. Returns True when
    drained, False on timeout (the drain flag stays set either way)."""
    deadline = time.monotonic() + timeout
    while True:
        remaining = live_allocations_on(storage, gpu_index, limits_writer)
        if not remaining:
            return True
        if progress is not None:
            progress(remaining)
        if time.monotonic() >= deadline:
            return False
        time.sleep(min(poll_interval, max(0.0, deadline - time.monotonic())))
