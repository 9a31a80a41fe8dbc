"""elastic_gpu_agent_amd — MI355X-native Kubernetes fractional-GPU agent.

A brand-new framework with the capabilities of elastic-ai/elastic-gpu-agent,
re-designed for AMD Instinct MI355X (gfx950, CDNA4):

- kubelet device-plugin gRPC surface (``elasticgpu.io/gpu-core`` percent units,
  ``elasticgpu.io/gpu-memory`` MiB units) — same resource names, pod-annotation
  scheme and on-disk allocation-state format as the reference
  (ref: pkg/plugins/gpushare.go, pkg/types/device.go).
- gfx950 device enumeration through **libamd_smi** (no NVML, no fallback),
  including the xGMI link table used for topology-aware preferred allocation.
- per-allocation device materialization targeting ``/dev/dri/renderD*`` and
  ``/dev/kfd`` (not /dev/nvidiaN), and an OCI prestart hook that injects those
  nodes directly (no wrapped vendor toolkit).
- a hand-written HSA interposer (C++ tools-lib) enforcing per-container CU
  masks (XCD-aware, ``hsa_amd_queue_cu_set_mask``) and HBM quotas against the
  288 GB of HBM3E per GPU, verified by gfx950 HIP census kernels.
"""

__version__ = "0.1.0"
