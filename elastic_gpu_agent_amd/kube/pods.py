"""Minimal pod model — just the fields the agent consumes."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional

from .. import consts


@dataclass
class Pod:
    namespace: str
    name: str
    uid: str = ""
    node_name: str = ""
    annotations: Dict[str, str] = field(default_factory=dict)
    phase: str = "Pending"
    deleted: bool = False

    @staticmethod
    def from_api_obj(obj: dict) -> "Pod":
        meta = obj.get("metadata", {})
        return Pod(
            namespace=meta.get("namespace", ""),
            name=meta.get("name", ""),
            uid=meta.get("uid", ""),
            node_name=obj.get("spec", {}).get("nodeName", ""),
            annotations=meta.get("annotations") or {},
            phase=obj.get("status", {}).get("phase", ""),
            deleted=bool(meta.get("deletionTimestamp")),
        )

    def is_assumed(self) -> bool:
        """Whether the elastic-gpu-scheduler has bound this pod
        (annotation contract, ref: pkg/common/const.go:7)."""
        return self.annotations.get(consts.ELASTIC_GPU_ASSUMED_ANNOTATION) == "true"

    def qos_class(self) -> Optional[str]:
        v = self.annotations.get(consts.ELASTIC_GPU_QOS_ANNOTATION)
        return v if v in ("low", "normal", "high") else None

    def container_gpu_indexes(self, container: str) -> Optional[str]:
        """Raw value of the per-container binding annotation
        (comma-separated GPU indexes, ref: pkg/plugins/gpushare.go:107-125)."""
        return self.annotations.get(consts.ELASTIC_GPU_CONTAINER_ANNOTATION % container)
