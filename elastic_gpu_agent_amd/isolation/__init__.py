"""Isolation control plane: CU-mask bookkeeping + per-allocation limits files.

The data plane is the hand-written HSA interposer (native/egpu_shim.cpp,
loaded into containers via HSA_TOOLS_LIB) which applies CU masks with
``hsa_amd_queue_cu_set_mask`` and enforces HBM quotas by intercepting HSA
memory-pool allocation. This module is the agent-side half:

- ``CUMaskAllocator``: picks disjoint XCD-round-robin CU sets per GPU for
  live fractional allocations (persisted in the storage aux table so masks
  survive agent restarts and are reclaimed by GC).
- ``LimitsWriter``: writes the per-allocation ``<hash>.json`` limits file the
  shim reads inside the container (mounted by the Allocate response).

The reference has no open-source equivalent (its QoS lived in the closed qGPU
driver) — this layer is MI355X-scoped, not a port.
"""
from __future__ import annotations

import json
import os
import threading
from typing import Dict, List, Optional, Tuple

from .. import consts
from ..types import GPUDevice
from .cumask import mask_hex, mask_words_from_cus, parse_mask_hex

AUX_MASK_PREFIX = "mask/"  # aux key: mask/<alloc_hash> -> json record


class CUMaskAllocator:
    """Allocates disjoint CU sets (spread across XCDs) per physical GPU.

    Occupancy is reconstructed from the storage aux table on startup, so a
    restarted agent keeps honoring masks of running pods (the Restore path
    the reference declared but never implemented — pkg/manager/manager.go:20).
    """

    def __init__(self, storage, devices: List[GPUDevice]):
        self._storage = storage
        self._devices = {d.index: d for d in devices}
        self._lock = threading.Lock()
        # In-memory occupancy cache, rebuilt from the aux table at startup
        # (restart safety) and maintained on allocate/release: per-allocation
        # work must not scan every live record (O(n²) under churn).
        self._used: Dict[int, set] = {}
        self._by_hash: Dict[str, tuple] = {}  # hash -> (gpu_index, set_of_cus)
        for key, val in storage.aux_items(AUX_MASK_PREFIX):
            rec = json.loads(val)
            cus = self._mask_cus(rec["cu_mask"])
            gpu = rec.get("gpu_index")
            self._by_hash[key[len(AUX_MASK_PREFIX):]] = (gpu, cus)
            self._used.setdefault(gpu, set()).update(cus)

    @staticmethod
    def _mask_cus(hexmask: str) -> set:
        cus = set()
        for w_i, w in enumerate(parse_mask_hex(hexmask)):
            for b in range(32):
                if w >> b & 1:
                    cus.add(w_i * 32 + b)
        return cus

    # ---- occupancy ----
    def _live_cus(self, gpu_index: int) -> set:
        return self._used.setdefault(gpu_index, set())

    def allocate(self, alloc_hash: str, gpu_index: int, percent: int) -> Tuple[str, int]:
        """Pick a CU set of ``percent``% of the GPU, disjoint from live
        allocations when capacity allows (oversubscription falls back to
        overlapping masks — documented QoS mode). Returns (mask_hex, n_cus)."""
        dev = self._devices.get(gpu_index)
        total = dev.cu_count if dev else consts.GFX950_CU_COUNT
        xcds = dev.xcd_count if dev else consts.GFX950_XCD_COUNT
        per_xcd = total // xcds
        from .cumask import cu_count_for_percent

        n = cu_count_for_percent(percent, total)
        pairs_per_xcd = per_xcd // 2
        with self._lock:
            used = self._live_cus(gpu_index)
            # ROCr CU masks have pair granularity: allocate whole CU pairs.
            # Water-filling round-robin over the XCDs' FREE pairs: as long as
            # free capacity exists anywhere, a new allocation never overlaps
            # (even when earlier pods fragmented some XCDs), while staying as
            # XCD-balanced as the free space allows. Only genuine
            # oversubscription falls back to overlapping pairs.
            want = (n + 1) // 2
            free_by_xcd = [
                [
                    xcd * per_xcd + 2 * p
                    for p in range(pairs_per_xcd)
                    if xcd * per_xcd + 2 * p not in used
                    and xcd * per_xcd + 2 * p + 1 not in used
                ]
                for xcd in range(xcds)
            ]
            taken: List[int] = []
            while len(taken) < want and any(free_by_xcd):
                progress = False
                for xcd in range(xcds):
                    if len(taken) >= want:
                        break
                    if free_by_xcd[xcd]:
                        taken.append(free_by_xcd[xcd].pop(0))
                        progress = True
                if not progress:
                    break
            if len(taken) < want:
                # oversubscribed: overlap already-used pairs, round-robin
                # across XCDs so the overlap is spread too
                for p in range(pairs_per_xcd):
                    for xcd in range(xcds):
                        if len(taken) >= want:
                            break
                        cu0 = xcd * per_xcd + 2 * p
                        if cu0 not in taken:
                            taken.append(cu0)
                    if len(taken) >= want:
                        break
            cus: List[int] = []
            for cu0 in taken:
                cus.extend((cu0, cu0 + 1))
            n_eff = len(cus)  # pair rounding may add one CU over the ask
            words = mask_words_from_cus(cus, total)
            hexmask = mask_hex(words)
            self._storage.aux_set(
                AUX_MASK_PREFIX + alloc_hash,
                json.dumps(
                    {
                        "gpu_index": gpu_index,
                        "cu_mask": hexmask,
                        "cu_count": n_eff,
                        "percent": percent,
                    }
                ),
            )
            cu_set = set(cus)
            old = self._by_hash.pop(alloc_hash, None)
            if old is not None:  # re-allocation of the same hash
                self._used.setdefault(old[0], set()).difference_update(old[1])
            self._by_hash[alloc_hash] = (gpu_index, cu_set)
            self._used.setdefault(gpu_index, set()).update(cu_set)
        return hexmask, n_eff

    def _release_cached(self, alloc_hash: str) -> None:
        entry = self._by_hash.pop(alloc_hash, None)
        if entry is None:
            return
        gpu_index, cus = entry
        used = self._used.setdefault(gpu_index, set())
        # only remove CUs not still claimed by another live mask
        still = set()
        for g, c in self._by_hash.values():
            if g == gpu_index:
                still |= c
        used.difference_update(cus - still)

    def release(self, alloc_hash: str) -> None:
        with self._lock:
            self._release_cached(alloc_hash)
        self._storage.aux_delete(AUX_MASK_PREFIX + alloc_hash)

    def release_many(self, hashes) -> None:
        """Batch form for GC: one cache pass + one storage transaction."""
        hashes = list(hashes)
        with self._lock:
            for h in hashes:
                self._release_cached(h)
        self._storage.aux_delete_many(AUX_MASK_PREFIX + h for h in hashes)

    def get(self, alloc_hash: str) -> Optional[dict]:
        raw = self._storage.aux_get(AUX_MASK_PREFIX + alloc_hash)
        return json.loads(raw) if raw else None


class LimitsWriter:
    """Per-allocation limits files consumed by the HSA shim in-container.

    Layout: ``<limits_dir>/<hash>.json`` on the host, mounted read-only at
    ``/etc/egpu/limits-<kind>.json`` in the container. Written (empty) at
    Allocate time so kubelet can mount it, finalized at PreStart once the
    physical GPU binding is known."""

    def __init__(self, limits_dir: str):
        self.limits_dir = limits_dir
        os.makedirs(limits_dir, exist_ok=True)

    def host_path(self, alloc_hash: str) -> str:
        return os.path.join(self.limits_dir, f"{alloc_hash}.json")

    @staticmethod
    def container_path(kind: str) -> str:
        return f"/etc/egpu/limits-{kind}.json"

    def touch(self, alloc_hash: str) -> str:
        p = self.host_path(alloc_hash)
        if not os.path.exists(p):
            self._atomic_write(p, {})
        return p

    def finalize(
        self,
        alloc_hash: str,
        gpu_indexes: List[int],
        devices: List[GPUDevice],
        cu_mask: Optional[str] = None,
        cu_count: Optional[int] = None,
        mem_limit_bytes: Optional[int] = None,
        priority: Optional[str] = None,
    ) -> None:
        dev_by_idx = {d.index: d for d in devices}
        rec: Dict = {
            "version": 1,
            "gpu_indexes": gpu_indexes,
            "render_minors": [
                dev_by_idx[i].drm_render_minor for i in gpu_indexes if i in dev_by_idx
            ],
            "uuids": [dev_by_idx[i].uuid for i in gpu_indexes if i in dev_by_idx],
        }
        if cu_mask is not None:
            rec["cu_mask"] = cu_mask
            rec["cu_count"] = cu_count
        if mem_limit_bytes is not None:
            rec["mem_limit_bytes"] = mem_limit_bytes
        if priority is not None:
            rec["priority"] = priority
        self._atomic_write(self.host_path(alloc_hash), rec)

    def delete(self, alloc_hash: str) -> None:
        try:
            os.unlink(self.host_path(alloc_hash))
        except FileNotFoundError:
            pass

    def read(self, alloc_hash: str) -> dict:
        with open(self.host_path(alloc_hash)) as f:
            return json.load(f)

    @staticmethod
    def _atomic_write(path: str, obj: dict) -> None:
        # unique temp name: concurrent writers of the SAME allocation (e.g. a
        # kubelet retry racing the first attempt) must not clobber each
        # other's temp file (found by tools/soak.py)
        import tempfile

        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(path), suffix=".tmp")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(obj, f)
            os.replace(tmp, path)
        except BaseException:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise
