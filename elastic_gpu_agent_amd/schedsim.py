"""Scheduler simulator: the elastic-gpu-scheduler's node-local decision,
re-implemented against this agent's GetPreferredAllocation.

The real ecosystem splits responsibilities: the external elastic-gpu-scheduler
picks GPUs cluster-wide and writes the pod annotations; the agent binds.
This module provides the node-local half of that decision for tests, benches
and kind-cluster demos (BASELINE config #1): given a pod's resource request,
it asks the agent's GetPreferredAllocation for the best device IDs (the
xGMI/NUMA-aware packing the reference stubbed out) and produces exactly the
annotations the agent's PreStart expects.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from . import consts
from .topology import gpu_index_of, prefer_allocation


@dataclass
class SimScheduler:
    """Tracks free units per GPU on one node and places pods."""

    devices: List  # types.GPUDevice
    mem_unit_mib: int = 1024
    # free unit IDs per resource
    free_core: Dict[str, bool] = field(default_factory=dict)
    free_mem: Dict[str, bool] = field(default_factory=dict)

    def __post_init__(self):
        for gpu in self.devices:
            for slot in range(consts.GPU_PERCENT_EACH_CARD):
                self.free_core[f"{gpu.index}-{slot:02d}"] = True
            for slot in range(gpu.memory_mib // self.mem_unit_mib):
                self.free_mem[f"{gpu.index}-{slot:06d}"] = True

    # kubelet removes Unhealthy devices from the allocatable pool; mirror
    # that so drains (per-GPU Unhealthy re-advertisement) stop placements
    unhealthy: set = field(default_factory=set)

    def _available(self, resource: str) -> List[str]:
        pool = self.free_core if resource == consts.RESOURCE_GPU_CORE else self.free_mem
        return [k for k, free in pool.items() if free and k not in self.unhealthy]

    def sync_health(self, *plugins) -> None:
        """Ingest ListAndWatch health from agent plugin(s), as kubelet would."""
        for plugin in plugins:
            for dev in plugin.list_devices(True):
                if dev["health"] == consts.HEALTHY:
                    self.unhealthy.discard(dev["ID"])
                else:
                    self.unhealthy.add(dev["ID"])

    def place(
        self,
        container: str,
        core_units: int = 0,
        memory_units: int = 0,
        preferred_fn=None,
    ) -> Optional[dict]:
        """Pick device IDs for one container; returns
        {"annotations": {...}, "core_ids": [...], "memory_ids": [...]} or
        None if the node cannot fit the request.

        ``preferred_fn(resource, available, size) -> ids`` lets callers route
        through the agent's live GetPreferredAllocation RPC; the default uses
        the same topology logic in-process."""
        annotations = {consts.ELASTIC_GPU_ASSUMED_ANNOTATION: "true"}
        picked_core: List[str] = []
        picked_mem: List[str] = []

        def pick(resource, size):
            avail = self._available(resource)
            if len(avail) < size:
                return None
            if preferred_fn is not None:
                ids = preferred_fn(resource, avail, size)
            else:
                single = (resource == consts.RESOURCE_GPU_MEMORY
                          or size <= consts.GPU_PERCENT_EACH_CARD)
                ids = prefer_allocation(avail, [], size, self.devices,
                                        single_gpu=single)
            return ids if len(ids) == size else None

        if core_units:
            picked_core = pick(consts.RESOURCE_GPU_CORE, core_units)
            if picked_core is None:
                return None
        if memory_units:
            picked_mem = pick(consts.RESOURCE_GPU_MEMORY, memory_units)
            if picked_mem is None:
                return None

        # GPU indexes for the container annotation: core allocation order
        # (one index per started 100 units), or the memory allocation's GPU
        indexes: List[int] = []
        if picked_core:
            seen = []
            for did in picked_core:
                g = gpu_index_of(did)
                if g not in seen:
                    seen.append(g)
            indexes = seen
        elif picked_mem:
            indexes = [gpu_index_of(picked_mem[0])]
        annotations[consts.ELASTIC_GPU_CONTAINER_ANNOTATION % container] = ",".join(
            str(i) for i in indexes
        )

        for did in picked_core:
            self.free_core[did] = False
        for did in picked_mem:
            self.free_mem[did] = False
        return {
            "annotations": annotations,
            "core_ids": picked_core,
            "memory_ids": picked_mem,
            "gpu_indexes": indexes,
        }

    def release(self, core_ids: List[str] = (), memory_ids: List[str] = ()) -> None:
        for did in core_ids:
            self.free_core[did] = True
        for did in memory_ids:
            self.free_mem[did] = True
