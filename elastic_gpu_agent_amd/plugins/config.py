"""Shared dependency bundle for the device plugins
(role of the reference's GPUPluginConfig, ref: pkg/plugins/base.go:32-50)."""
from __future__ import annotations

import dataclasses
from typing import Optional

from .. import consts


@dataclasses.dataclass
class AgentPaths:
    """Host-side paths the agent writes / containers consume."""

    # -- agent-view paths (how THIS process reaches the host filesystem;
    #    inside the DaemonSet the host is mounted under /host) --
    dev_root: str = consts.HOST_DEV_ROOT  # where per-alloc symlinks live (host /dev)
    plugin_dir: str = consts.DEVICE_PLUGIN_PATH
    kubelet_socket: Optional[str] = None  # default: <plugin_dir>/kubelet.sock
    podresources_socket: str = consts.POD_RESOURCES_SOCKET
    limits_dir: str = "/host/var/lib/egpu/limits"
    state_dir: str = "/host/var/lib/egpu"  # hook-recorded pids live here
    # -- advertise-view paths (what kubelet/the runtime resolve ON THE HOST;
    #    Allocate responses must carry these, never the /host/... view) --
    limits_dir_host: Optional[str] = "/var/lib/egpu/limits"  # None = same as limits_dir
    # host path of the HSA shim library mounted into containers; None disables
    # isolation env injection entirely.
    shim_host_path: Optional[str] = "/opt/egpu/libegpu_shim.so"
    shim_container_path: str = "/opt/egpu/libegpu_shim.so"

    def limits_host_view(self, filename: str) -> str:
        import os as _os

        base = self.limits_dir_host if self.limits_dir_host is not None else self.limits_dir
        return _os.path.join(base, filename)


@dataclasses.dataclass
class PluginOptions:
    mem_unit_mib: int = 1  # MiB per gpu-memory unit (reference contract: 1)
    isolation: bool = True  # inject HSA shim + limits into fractional pods
    health_refresh_seconds: float = 30.0


@dataclasses.dataclass
class GPUPluginConfig:
    operator: object  # operator.GPUOperator
    storage: object  # storage.Storage
    sitter: object  # kube.sitter.Sitter
    core_locator: object  # kube.locator.DeviceLocator
    memory_locator: object  # kube.locator.DeviceLocator
    paths: AgentPaths = dataclasses.field(default_factory=AgentPaths)
    options: PluginOptions = dataclasses.field(default_factory=PluginOptions)
    limits: Optional[object] = None  # isolation.LimitsWriter
    cumask: Optional[object] = None  # isolation.CUMaskAllocator
    # optional callable(namespace, pod_name, reason, message) emitting a k8s
    # Event; must be non-blocking best-effort
    event_sink: Optional[object] = None
