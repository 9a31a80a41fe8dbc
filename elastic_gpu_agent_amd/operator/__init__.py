"""GPU operator layer: device enumeration backends + device-node materialization.

Mirrors the reference's GPUOperator surface {Devices; Create; Delete; Check}
(ref: pkg/operator/base.go:9-14) with MI355X-native semantics:

- enumeration comes from **libamd_smi** (AmdSmiBackend) or a synthetic gfx950
  fleet (FakeBackend) — never NVML, no fallback chain: a GPU node that cannot
  load libamd_smi fails loudly.
- per-allocation device nodes are symlinks
  ``<dev_root>/elastic-gpu-<id>``    → ``/dev/dri/renderD<minor>`` and
  ``<dev_root>/elastic-gpuctl-<id>`` → ``/dev/kfd``
  (ref kept the same naming against /dev/nvidiaN: pkg/operator/gpushare.go:9-16).
"""
from __future__ import annotations

import abc
import os
from typing import Dict, List, Optional

from .. import consts
from ..types import GPUDevice


class GPUBackend(abc.ABC):
    """Enumerates physical (or synthetic) GPUs."""

    @abc.abstractmethod
    def devices(self) -> List[GPUDevice]:
        ...


class GPUOperator:
    """Backend + symlink lifecycle for per-allocation device nodes."""

    def __init__(self, backend: GPUBackend, dev_root: str = consts.HOST_DEV_ROOT):
        self.backend = backend
        self.dev_root = dev_root
        self._cache: Optional[List[GPUDevice]] = None

    # ---- enumeration ----
    def devices(self, refresh: bool = False) -> List[GPUDevice]:
        if self._cache is None or refresh:
            self._cache = self.backend.devices()
        return self._cache

    def device_by_index(self, index: int) -> GPUDevice:
        for d in self.devices():
            if d.index == index:
                return d
        raise KeyError(f"no GPU with index {index}")

    # ---- materialization ----
    def _link_paths(self, alloc_id: str) -> Dict[str, str]:
        return {
            "gpu": os.path.join(self.dev_root, consts.ELASTIC_GPU_LINK_FMT % alloc_id),
            "ctl": os.path.join(self.dev_root, consts.ELASTIC_GPU_CTL_LINK_FMT % alloc_id),
        }

    def create(self, gpu_index: int, alloc_id: str) -> None:
        """Create the per-allocation symlinks for one GPU.

        The GPU link targets the DRM render node of the physical GPU; the ctl
        link targets /dev/kfd (the ROCm compute control node — the role
        /dev/nvidiactl plays in the reference)."""
        dev = self.device_by_index(gpu_index)
        paths = self._link_paths(alloc_id)
        os.makedirs(self.dev_root, exist_ok=True)
        self._force_symlink(consts.DRI_RENDER_FMT % dev.drm_render_minor, paths["gpu"])
        self._force_symlink(consts.KFD_PATH, paths["ctl"])

    def delete(self, gpu_index: int, alloc_id: str) -> None:
        """Remove the per-allocation symlinks. ``gpu_index`` may be -1 when the
        GPU is unknown (GC path, ref: pkg/plugins/base.go:281-293)."""
        for p in self._link_paths(alloc_id).values():
            try:
                os.unlink(p)
            except FileNotFoundError:
                pass

    def check(self, gpu_index: int, alloc_id: str) -> bool:
        paths = self._link_paths(alloc_id)
        return all(os.path.islink(p) for p in paths.values())

    @staticmethod
    def _force_symlink(target: str, link: str) -> None:
        # bounded retry: concurrent binds of the same device-set hash and GC
        # passes may create/unlink the same link (two pods requesting an
        # identical ID set share a hash by construction)
        for _ in range(5):
            try:
                os.symlink(target, link)
                return
            except FileExistsError:
                try:
                    if os.readlink(link) == target:
                        return
                except OSError:
                    pass
                try:
                    os.unlink(link)
                except FileNotFoundError:
                    pass
        raise OSError(f"could not materialize symlink {link} -> {target}")
