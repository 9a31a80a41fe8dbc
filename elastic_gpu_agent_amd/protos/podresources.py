"""kubelet podresources API v1alpha1 message specs + gRPC method name.

Field numbers follow the upstream Kubernetes proto contract
(k8s.io/kubelet podresources v1alpha1 api.proto; the reference carries the
same file in-tree at pkg/podresources/v1alpha1/api.proto:19-48).
"""
from .protowire import MessageSpec

ListPodResourcesRequest = MessageSpec("ListPodResourcesRequest", [])

ContainerDevices = MessageSpec(
    "ContainerDevices",
    [
        (1, "resource_name", "string", None, False),
        (2, "device_ids", "string", None, True),
    ],
)
ContainerResources = MessageSpec(
    "ContainerResources",
    [
        (1, "name", "string", None, False),
        (2, "devices", "message", ContainerDevices, True),
    ],
)
PodResources = MessageSpec(
    "PodResources",
    [
        (1, "name", "string", None, False),
        (2, "namespace", "string", None, False),
        (3, "containers", "message", ContainerResources, True),
    ],
)
ListPodResourcesResponse = MessageSpec(
    "ListPodResourcesResponse", [(1, "pod_resources", "message", PodResources, True)]
)

POD_RESOURCES_SERVICE = "v1alpha1.PodResourcesLister"
METHOD_LIST = f"/{POD_RESOURCES_SERVICE}/List"
