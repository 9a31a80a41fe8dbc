"""Synthetic gfx950 fleet for tests, the kind-cluster config and CPU benches.

The reference has no fake backend (its tests hit NVML for real — SURVEY §4);
this one models an 8×MI355X node: 256 CUs / 8 XCDs / 288 GiB HBM3E per GPU,
full xGMI mesh (7 point-to-point links per GPU on an 8-GPU node).
"""
from __future__ import annotations

import os
from typing import List, Optional

from .. import consts
from ..types import GPUDevice
from . import GPUBackend


class FakeBackend(GPUBackend):
    def __init__(self, count: Optional[int] = None, memory_bytes: int = consts.GFX950_HBM_BYTES):
        if count is None:
            count = int(os.environ.get("EGPU_FAKE_GPUS", "8"))
        self.count = count
        self.memory_bytes = memory_bytes

    def devices(self) -> List[GPUDevice]:
        devs = []
        for i in range(self.count):
            peers = tuple(j for j in range(self.count) if j != i)
            devs.append(
                GPUDevice(
                    uuid=f"GPU-fake-gfx950-{i:04d}",
                    index=i,
                    memory_bytes=self.memory_bytes,
                    drm_render_minor=128 + i,
                    drm_card=i,
                    cu_count=consts.GFX950_CU_COUNT,
                    xcd_count=consts.GFX950_XCD_COUNT,
                    numa_node=i // 4,  # 2 NUMA domains on a typical 8-GPU node
                    xgmi_peers=peers,
                    compute_partition="SPX",
                )
            )
        return devs
