"""Shared constants.

Keeps the reference's public contract values (ref: pkg/common/const.go:3-8,
vendor/elasticgpu.io/elastic-gpu/api/v1alpha1/types.go:105-112,
vendor/k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/constants.go:26-32) while
the device-node side is MI355X-native (/dev/kfd + /dev/dri/renderD*).
"""

# --- elasticgpu.io resource & annotation contract (unchanged from reference) ---
RESOURCE_GPU_CORE = "elasticgpu.io/gpu-core"
RESOURCE_GPU_MEMORY = "elasticgpu.io/gpu-memory"

# 100 "percent" units advertised per physical GPU for the core resource.
GPU_PERCENT_EACH_CARD = 100

# Pod annotations written by the elastic-gpu-scheduler and consumed here.
ELASTIC_GPU_ASSUMED_ANNOTATION = "elasticgpu.io/assumed"
ELASTIC_GPU_CONTAINER_ANNOTATION = "elasticgpu.io/container-%s"
# optional per-pod QoS class consumed by the HSA shim (queue priority):
ELASTIC_GPU_QOS_ANNOTATION = "elasticgpu.io/qos-class"  # low | normal | high

NODE_NAME_FIELD = "spec.nodeName"

# --- kubelet device-plugin API v1beta1 (upstream k8s contract) ---
DEVICE_PLUGIN_VERSION = "v1beta1"
DEVICE_PLUGIN_PATH = "/var/lib/kubelet/device-plugins/"
KUBELET_SOCKET = DEVICE_PLUGIN_PATH + "kubelet.sock"
HEALTHY = "Healthy"
UNHEALTHY = "Unhealthy"

CORE_SOCK_NAME = "elastic-gpushare-core.sock"
MEMORY_SOCK_NAME = "elastic-gpushare-mem.sock"

# --- kubelet podresources API v1alpha1 ---
POD_RESOURCES_PATH = "/var/lib/kubelet/pod-resources"
POD_RESOURCES_SOCKET = POD_RESOURCES_PATH + "/kubelet.sock"
POD_RESOURCES_MAX_SIZE = 1024 * 1024 * 16  # 16 MiB

# --- MI355X device-node materialization -------------------------------------
# Per-allocation symlinks live on the host /dev; names keep the reference's
# "elastic-gpu-<id>" scheme (ref: pkg/operator/gpushare.go:9-16) but the
# *targets* are the AMDGPU/KFD nodes.
HOST_DEV_ROOT = "/host/dev"
ELASTIC_GPU_LINK_FMT = "elastic-gpu-%s"  # -> /dev/dri/renderD<minor>
ELASTIC_GPU_CTL_LINK_FMT = "elastic-gpuctl-%s"  # -> /dev/kfd
KFD_PATH = "/dev/kfd"
DRI_RENDER_FMT = "/dev/dri/renderD%d"

# Env var set in Allocate responses; consumed by the OCI prestart hook.
GPU_ENV_KEY = "GPU"

# MI355X (gfx950) hardware shape used by the fake backend and the CU-mask math.
GFX950_CU_COUNT = 256
GFX950_XCD_COUNT = 8
GFX950_CU_PER_XCD = 32
GFX950_HBM_BYTES = 288 * 1024**3  # 288 GiB HBM3E
GFX950_XGMI_LINKS = 7  # point-to-point links per GPU on an 8-GPU node
