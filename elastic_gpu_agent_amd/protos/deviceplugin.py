"""kubelet device-plugin API v1beta1 message specs + gRPC method names.

Field numbers follow the upstream Kubernetes proto contract
(k8s.io/kubelet/pkg/apis/deviceplugin/v1beta1/api.proto; the reference vendors
the same file). Messages are plain dicts shaped by MessageSpec.
"""
from .protowire import MessageSpec

# ---- messages --------------------------------------------------------------
DevicePluginOptions = MessageSpec(
    "DevicePluginOptions",
    [
        (1, "pre_start_required", "bool", None, False),
        (2, "get_preferred_allocation_available", "bool", None, False),
    ],
)

RegisterRequest = MessageSpec(
    "RegisterRequest",
    [
        (1, "version", "string", None, False),
        (2, "endpoint", "string", None, False),
        (3, "resource_name", "string", None, False),
        (4, "options", "message", DevicePluginOptions, False),
    ],
)

Empty = MessageSpec("Empty", [])

NUMANode = MessageSpec("NUMANode", [(1, "ID", "int64", None, False)])
TopologyInfo = MessageSpec("TopologyInfo", [(1, "nodes", "message", NUMANode, True)])

Device = MessageSpec(
    "Device",
    [
        (1, "ID", "string", None, False),
        (2, "health", "string", None, False),
        (3, "topology", "message", TopologyInfo, False),
    ],
)

ListAndWatchResponse = MessageSpec(
    "ListAndWatchResponse", [(1, "devices", "message", Device, True)]
)

PreStartContainerRequest = MessageSpec(
    "PreStartContainerRequest", [(1, "devicesIDs", "string", None, True)]
)
PreStartContainerResponse = MessageSpec("PreStartContainerResponse", [])

ContainerPreferredAllocationRequest = MessageSpec(
    "ContainerPreferredAllocationRequest",
    [
        (1, "available_deviceIDs", "string", None, True),
        (2, "must_include_deviceIDs", "string", None, True),
        (3, "allocation_size", "int32", None, False),
    ],
)
PreferredAllocationRequest = MessageSpec(
    "PreferredAllocationRequest",
    [(1, "container_requests", "message", ContainerPreferredAllocationRequest, True)],
)
ContainerPreferredAllocationResponse = MessageSpec(
    "ContainerPreferredAllocationResponse", [(1, "deviceIDs", "string", None, True)]
)
PreferredAllocationResponse = MessageSpec(
    "PreferredAllocationResponse",
    [(1, "container_responses", "message", ContainerPreferredAllocationResponse, True)],
)

ContainerAllocateRequest = MessageSpec(
    "ContainerAllocateRequest", [(1, "devicesIDs", "string", None, True)]
)
AllocateRequest = MessageSpec(
    "AllocateRequest", [(1, "container_requests", "message", ContainerAllocateRequest, True)]
)

Mount = MessageSpec(
    "Mount",
    [
        (1, "container_path", "string", None, False),
        (2, "host_path", "string", None, False),
        (3, "read_only", "bool", None, False),
    ],
)
DeviceSpec = MessageSpec(
    "DeviceSpec",
    [
        (1, "container_path", "string", None, False),
        (2, "host_path", "string", None, False),
        (3, "permissions", "string", None, False),
    ],
)
ContainerAllocateResponse = MessageSpec(
    "ContainerAllocateResponse",
    [
        (1, "envs", "map_str_str", None, False),
        (2, "mounts", "message", Mount, True),
        (3, "devices", "message", DeviceSpec, True),
        (4, "annotations", "map_str_str", None, False),
    ],
)
AllocateResponse = MessageSpec(
    "AllocateResponse", [(1, "container_responses", "message", ContainerAllocateResponse, True)]
)

# ---- gRPC method names -----------------------------------------------------
REGISTRATION_SERVICE = "v1beta1.Registration"
DEVICE_PLUGIN_SERVICE = "v1beta1.DevicePlugin"

METHOD_REGISTER = f"/{REGISTRATION_SERVICE}/Register"
METHOD_GET_OPTIONS = f"/{DEVICE_PLUGIN_SERVICE}/GetDevicePluginOptions"
METHOD_LIST_AND_WATCH = f"/{DEVICE_PLUGIN_SERVICE}/ListAndWatch"
METHOD_GET_PREFERRED_ALLOCATION = f"/{DEVICE_PLUGIN_SERVICE}/GetPreferredAllocation"
METHOD_ALLOCATE = f"/{DEVICE_PLUGIN_SERVICE}/Allocate"
METHOD_PRE_START_CONTAINER = f"/{DEVICE_PLUGIN_SERVICE}/PreStartContainer"
