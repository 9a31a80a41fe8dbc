"""In-process podresources List server (gRPC over UDS).

Serves the kubelet podresources v1alpha1 List API from a settable table —
used by unit tests and the benchmark as the stand-in kubelet (the reference
carries an equivalent server shape in-tree: pkg/podresources/server.go:44-75).
"""
from __future__ import annotations

import threading
from typing import Dict, List

from .. import egrpc
from ..protos import podresources as pr


class PodResourcesServer:
    def __init__(self, socket_path: str, max_workers: int = 4):
        self._socket = socket_path
        self._lock = threading.Lock()
        # table: (ns, pod) -> {container: [(resource_name, [ids...]), ...]}
        self._table: Dict[tuple, Dict[str, List[tuple]]] = {}
        self._server = egrpc.Server()
        self._server.add_service(
            pr.POD_RESOURCES_SERVICE,
            {
                "List": egrpc.unary_unary(
                    self._list,
                    request_deserializer=pr.ListPodResourcesRequest.decode,
                    response_serializer=pr.ListPodResourcesResponse.encode,
                )
            },
        )
        self._server.bind_unix(socket_path)

    def set_assignment(
        self, namespace: str, pod: str, container: str, resource_name: str, device_ids: List[str]
    ) -> None:
        with self._lock:
            self._table.setdefault((namespace, pod), {}).setdefault(container, []).append(
                (resource_name, list(device_ids))
            )

    def remove_pod(self, namespace: str, pod: str) -> None:
        with self._lock:
            self._table.pop((namespace, pod), None)

    def _list(self, request, context):
        with self._lock:
            pods = []
            for (ns, name), containers in self._table.items():
                pods.append(
                    {
                        "name": name,
                        "namespace": ns,
                        "containers": [
                            {
                                "name": cname,
                                "devices": [
                                    {"resource_name": rn, "device_ids": ids}
                                    for rn, ids in entries
                                ],
                            }
                            for cname, entries in containers.items()
                        ],
                    }
                )
        return {"pod_resources": pods}

    def start(self) -> None:
        self._server.start()

    def stop(self) -> None:
        self._server.stop(grace=0.2)
