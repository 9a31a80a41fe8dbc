"""Real MI355X enumeration through libamd_smi (native pybind11 binding).

The reference enumerates through NVML cgo (ref: pkg/operator/base.go:19-75);
here the native binding ``elastic_gpu_agent_amd._amdsmi`` (C++, linked against
/opt/rocm/lib/libamd_smi.so) returns per-GPU: index (HIP enumeration order),
UUID, VRAM bytes, DRM render minor, CU count, NUMA node, compute-partition
mode and the xGMI peer table. There is deliberately NO fallback: a GPU node
where the binding or libamd_smi is missing raises ImportError/RuntimeError
instead of silently degrading.
"""
from __future__ import annotations

from typing import List

from ..types import GPUDevice
from . import GPUBackend


class AmdSmiBackend(GPUBackend):
    def __init__(self):
        try:
            from elastic_gpu_agent_amd import _amdsmi  # native extension, built in-tree
        except ImportError as e:
            raise ImportError(
                "elastic_gpu_agent_amd._amdsmi native extension not built; "
                "run `python -m elastic_gpu_agent_amd.native.build` (requires hipcc/g++ "
                "and /opt/rocm/lib/libamd_smi.so). There is no non-amdsmi fallback."
            ) from e
        self._smi = _amdsmi

    def devices(self) -> List[GPUDevice]:
        raw = self._smi.enumerate_gpus()
        devs = []
        for d in raw:
            devs.append(
                GPUDevice(
                    uuid=d["uuid"],
                    index=d["index"],
                    memory_bytes=d["memory_bytes"],
                    drm_render_minor=d["drm_render"],
                    drm_card=d["drm_card"],
                    cu_count=d["cu_count"],
                    xcd_count=d.get("xcd_count", 8),
                    numa_node=d.get("numa_node", 0),
                    xgmi_peers=tuple(d.get("xgmi_peers", ())),
                    compute_partition=d.get("compute_partition", "SPX"),
                )
            )
        devs.sort(key=lambda g: g.index)
        return devs
