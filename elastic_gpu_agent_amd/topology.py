"""xGMI-topology-aware device selection.

The reference stubs GetPreferredAllocation (ref: pkg/plugins/base.go:94-96);
on an 8×MI355X node it matters: xGMI is 7 point-to-point links per GPU
(~153 GB/s each), so co-scheduled fractional pods should land on (a) the same
GPU when they fit, (b) xGMI-peer GPUs otherwise — never across a PCIe hop.

Device IDs encode their GPU as ``<gpu_index>-<slot>`` (core: 100 slots/GPU,
ref: pkg/plugins/gpushare.go:24-32; memory: one slot per unit). Selection
works on whole GPUs first, then slices IDs.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, List, Sequence

from .types import GPUDevice


def gpu_index_of(device_id: str) -> int:
    return int(device_id.split("-", 1)[0])


def group_by_gpu(device_ids: Sequence[str]) -> Dict[int, List[str]]:
    groups: Dict[int, List[str]] = defaultdict(list)
    for did in device_ids:
        groups[gpu_index_of(did)].append(did)
    return groups


def xgmi_score(gpus: Sequence[int], devices: Dict[int, GPUDevice]) -> int:
    """Number of directly-linked (xGMI-peer) pairs inside the candidate set."""
    score = 0
    for i, a in enumerate(gpus):
        peers = set(devices[a].xgmi_peers) if a in devices else set()
        for b in gpus[i + 1 :]:
            if b in peers:
                score += 1
    return score


def prefer_allocation(
    available: Sequence[str],
    must_include: Sequence[str],
    size: int,
    devices: List[GPUDevice],
    single_gpu: bool = False,
) -> List[str]:
    """Pick ``size`` device IDs from ``available`` ⊇ ``must_include``.

    Strategy (fractional-first, then topology):
    1. honor must_include;
    2. fill from the GPU(s) already used by must_include, then from the
       fewest-fragment GPUs (pack fractions tightly → whole GPUs stay free);
    3. when several GPUs are needed, choose the set maximizing pairwise xGMI
       links, tie-broken by NUMA-node co-location and lower index.

    ``single_gpu=True`` restricts the pick to ONE GPU (required for every
    memory allocation and for fractional core requests — PreStart binds
    those to exactly one physical GPU); when no single GPU has ``size``
    units free the pick is truncated, which callers treat as "does not
    fit" rather than silently producing an unbindable spanning set.
    """
    dev_by_idx = {d.index: d for d in devices}
    chosen: List[str] = list(must_include)
    chosen_set = set(chosen)
    remaining = size - len(chosen)
    if remaining <= 0:
        return chosen[:size]

    groups = group_by_gpu([d for d in available if d not in chosen_set])
    for idx in groups:
        groups[idx].sort()

    # GPUs already touched by must_include come first.
    touched = {gpu_index_of(d) for d in chosen}

    def gpu_order() -> List[int]:
        idxs = list(groups.keys())

        def key(g: int):
            # prefer: already-touched GPU; then most-loaded (fewest free slots —
            # tight packing); then xGMI adjacency to touched set; then NUMA.
            adj = 0
            if g in dev_by_idx:
                peers = set(dev_by_idx[g].xgmi_peers)
                adj = sum(1 for t in touched if t in peers)
            numa_match = 0
            if touched and g in dev_by_idx:
                numas = {dev_by_idx[t].numa_node for t in touched if t in dev_by_idx}
                numa_match = 1 if dev_by_idx[g].numa_node in numas else 0
            return (
                0 if g in touched else 1,
                len(groups[g]),  # fewer available => more loaded => pack first
                -adj,
                -numa_match,
                g,
            )

        idxs.sort(key=key)
        return idxs

    if single_gpu:
        # all units must come from one GPU (the one must_include touched,
        # else the best-ranked GPU with enough room)
        if touched:
            g = next(iter(touched))
            pool = groups.get(g, [])
            chosen.extend(pool[:remaining])
            return chosen
        for g in gpu_order():
            if len(groups[g]) >= remaining:
                chosen.extend(groups[g][:remaining])
                return chosen
        return chosen  # nothing fits: short pick, caller rejects

    while remaining > 0 and groups:
        order = gpu_order()
        g = order[0]
        take = groups[g][:remaining]
        chosen.extend(take)
        remaining -= len(take)
        del groups[g]
        touched.add(g)

    return chosen
