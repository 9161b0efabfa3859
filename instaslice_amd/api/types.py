"""L0: CRD-shaped data model — the `Instaslice` kind, kept schema-compatible.

Mirrors the reference's api/v1alpha1/instaslice_types.go:
  - AllocationDetails (:37-50)   -> AllocationDetails here
  - PreparedDetails   (:53-62)   -> PreparedDetails here
  - InstasliceSpec    (:65-72)   -> InstasliceSpec here (allocations, prepared,
                                    placements catalog, gpu_uuids map)
  - status-string protocol `creating -> created -> ungated -> deleted`
    (instaslice_controller.go:255, instaslice_daemonset.go:219,
     instaslice_controller.go:164, :113) -> AllocationStatus constants.

All objects serialize to/from plain dicts (the store speaks JSON), with
apiVersion/kind/metadata envelopes shaped like real Kubernetes objects so a
cluster adapter can pass them through unchanged.
"""

from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from instaslice_amd import API_GROUP, API_VERSION


class AllocationStatus:
    """The status-string protocol between controller and daemonset
    (SURVEY.md §1), extended with an explicit failure state the reference
    lacks (it logs-and-ignores NVML errors, instaslice_daemonset.go:173-189):

        creating -> created -> ungated -> deleted
                 \\-> failed (daemonset could not realize; controller
                              removes the allocation and re-places)
    """

    CREATING = "creating"   # controller decided placement; daemonset must realize
    CREATED = "created"     # daemonset realized the partition
    UNGATED = "ungated"     # controller removed the pod's scheduling gate
    DELETED = "deleted"     # controller marked for teardown; daemonset must clean
    FAILED = "failed"       # daemonset hit a hard device error; needs re-place

    ALL = (CREATING, CREATED, UNGATED, DELETED, FAILED)


@dataclass
class AllocationDetails:
    """One pod's partition allocation (reference: instaslice_types.go:37-50).

    `start`/`size` are XCD ordinal coordinates: `start` = partition ordinal
    under the GPU's target mode x xcds-per-partition (i.e. first XCD index),
    `size` = XCDs per partition — byte-compatible in spirit with the
    reference's MIG slot {start,size}.
    """

    profile: str                 # e.g. "cpx-1x36"
    gpu_uuid: str                # parent physical GPU
    ordinal: int                 # partition ordinal under the target mode
    start: int                   # first XCD index (ordinal * xcds)
    size: int                    # XCDs per partition
    pod_uuid: str
    pod_name: str
    namespace: str
    nodename: str
    allocation_status: str = AllocationStatus.CREATING
    compute_mode: str = ""       # target compute mode, e.g. "CPX"
    memory_mode: str = ""        # target memory mode, e.g. "NPS4"
    group: str = ""              # gang label (org.instaslice/group annotation)
    priority: int = 0            # org.instaslice/priority (preemption rank)

    def to_dict(self) -> dict:
        return {
            "profile": self.profile,
            "gpuUUID": self.gpu_uuid,
            "ordinal": self.ordinal,
            "start": self.start,
            "size": self.size,
            "podUUID": self.pod_uuid,
            "podName": self.pod_name,
            "namespace": self.namespace,
            "nodename": self.nodename,
            "allocationStatus": self.allocation_status,
            "computeMode": self.compute_mode,
            "memoryMode": self.memory_mode,
            "group": self.group,
            "priority": self.priority,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "AllocationDetails":
        return cls(
            profile=d["profile"],
            gpu_uuid=d["gpuUUID"],
            ordinal=d["ordinal"],
            start=d["start"],
            size=d["size"],
            pod_uuid=d["podUUID"],
            pod_name=d["podName"],
            namespace=d["namespace"],
            nodename=d["nodename"],
            allocation_status=d.get("allocationStatus", AllocationStatus.CREATING),
            compute_mode=d.get("computeMode", ""),
            memory_mode=d.get("memoryMode", ""),
            group=d.get("group", ""),
            priority=int(d.get("priority", 0)),
        )


@dataclass
class PreparedDetails:
    """A realized partition (reference: instaslice_types.go:53-62).

    Keyed in the spec by the partition's *device UUID* (the sub-device the
    workload will see via ROCR_VISIBLE_DEVICES), as the reference keys
    Prepared by MIG UUID.
    """

    parent_gpu_uuid: str
    ordinal: int
    compute_mode: str
    memory_mode: str
    xcds: int
    memory_gb: int
    pod_uuid: str
    # hip device index of the partition on its node (for in-process payloads)
    device_index: int = 0

    def to_dict(self) -> dict:
        return {
            "parentGpuUUID": self.parent_gpu_uuid,
            "ordinal": self.ordinal,
            "computeMode": self.compute_mode,
            "memoryMode": self.memory_mode,
            "xcds": self.xcds,
            "memoryGB": self.memory_gb,
            "podUUID": self.pod_uuid,
            "deviceIndex": self.device_index,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "PreparedDetails":
        return cls(
            parent_gpu_uuid=d["parentGpuUUID"],
            ordinal=d["ordinal"],
            compute_mode=d["computeMode"],
            memory_mode=d["memoryMode"],
            xcds=d["xcds"],
            memory_gb=d["memoryGB"],
            pod_uuid=d["podUUID"],
            device_index=d.get("deviceIndex", 0),
        )


@dataclass
class GpuStatus:
    """Per-GPU live mode/occupancy view persisted in the CR so the controller
    can plan mode transitions without talking to the device layer. The
    reference has no analog (MIG state is implicit in Prepared); AMD's
    whole-GPU mode semantics make this first-class."""

    uuid: str
    model: str
    memory_gb: int
    compute_mode: str = "SPX"
    memory_mode: str = "NPS1"
    # partition ordinals currently handed to pods (occupied)
    used_ordinals: List[int] = field(default_factory=list)

    def to_dict(self) -> dict:
        return {
            "uuid": self.uuid,
            "model": self.model,
            "memoryGB": self.memory_gb,
            "computeMode": self.compute_mode,
            "memoryMode": self.memory_mode,
            "usedOrdinals": sorted(self.used_ordinals),
        }

    @classmethod
    def from_dict(cls, d: dict) -> "GpuStatus":
        return cls(
            uuid=d["uuid"],
            model=d["model"],
            memory_gb=d["memoryGB"],
            compute_mode=d.get("computeMode", "SPX"),
            memory_mode=d.get("memoryMode", "NPS1"),
            used_ordinals=list(d.get("usedOrdinals", [])),
        )


def new_instaslice(node_name: str) -> dict:
    """Empty Instaslice CR for a node (reference: createInstasliceResource
    path, instaslice_daemonset.go:555-586)."""
    return {
        "apiVersion": f"{API_GROUP}/{API_VERSION}",
        "kind": "Instaslice",
        "metadata": {"name": node_name, "namespace": "instaslice-system"},
        "spec": {
            # gpuUUID -> model name (reference: MigGPUUUID, instaslice_types.go:66)
            "gpuUuids": {},
            # per-GPU mode/occupancy (AMD-specific; see GpuStatus)
            "gpus": {},
            # discovered profile catalog (reference: Migplacement, :71)
            "placements": {},
            # podUUID -> AllocationDetails (reference: Allocations, :68)
            "allocations": {},
            # partitionUUID -> PreparedDetails (reference: Prepared, :69)
            "prepared": {},
        },
        "status": {"processed": "false"},
    }


def new_pod(
    name: str,
    namespace: str = "default",
    profile: Optional[str] = None,
    uid: Optional[str] = None,
    node_selector: Optional[Dict[str, str]] = None,
    gated: bool = True,
    group: Optional[str] = None,
    labels: Optional[Dict[str, str]] = None,
    priority: int = 0,
) -> dict:
    """Synthetic gated pod following the reference's consumer contract
    (samples/test-pod.yaml:1-21): scheduling gate + finalizer + profile limit
    + pod-named extended resource + envFrom pod-named ConfigMap."""
    from instaslice_amd import FINALIZER_NAME, GATE_NAME, POD_RESOURCE_PREFIX, RESOURCE_PREFIX
    from instaslice_amd.utils import new_uid

    limits: Dict[str, object] = {}
    if profile:
        limits[RESOURCE_PREFIX + profile] = 1
        limits[POD_RESOURCE_PREFIX + name] = 1
    pod = {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": name,
            "namespace": namespace,
            "uid": uid or new_uid(),
            "finalizers": [FINALIZER_NAME],
            "deletionTimestamp": None,
            "annotations": {
                **({"org.instaslice/group": group} if group else {}),
                **({"org.instaslice/priority": str(priority)} if priority else {}),
            },
            "labels": labels or {},
        },
        "spec": {
            "schedulingGates": [{"name": GATE_NAME}] if gated else [],
            "nodeSelector": node_selector or {},
            "containers": [
                {
                    "name": "workload",
                    "image": "instaslice-payload",
                    "resources": {"limits": limits},
                    "envFrom": [{"configMapRef": {"name": name}}],
                }
            ],
        },
        "status": {
            "phase": "Pending",
            "conditions": (
                [{"type": "PodScheduled", "status": "False",
                  "message": "pod is blocked on scheduling gates"}]
                if gated else []
            ),
        },
    }
    return pod


def timestamp_epoch(ts) -> float:
    """Normalize a metadata timestamp to epoch seconds.

    MemStore/netstore/stored write epoch floats, but a real Kubernetes API
    server (the K8sStore backend) returns RFC3339 strings such as
    "2026-09-14T12:00:00Z" — grace-period math must accept both
    (advisor finding r1: float() on the string crash-looped teardown on
    the k8s adapter)."""
    if isinstance(ts, (int, float)):
        return float(ts)
    s = str(ts).strip()
    try:
        return float(s)
    except ValueError:
        pass
    from datetime import datetime, timezone

    if s.endswith(("Z", "z")):
        s = s[:-1] + "+00:00"
    dt = datetime.fromisoformat(s)
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=timezone.utc)
    return dt.timestamp()


def pod_is_gated(pod: dict) -> bool:
    """reference: checkIfPodGated, instaslice_controller.go:386-395."""
    from instaslice_amd import GATE_NAME

    gates = pod.get("spec", {}).get("schedulingGates", []) or []
    if not any(g.get("name") == GATE_NAME for g in gates):
        return False
    status = pod.get("status", {})
    if status.get("phase") != "Pending":
        return False
    conds = status.get("conditions", []) or []
    return bool(conds) and "blocked" in (conds[0].get("message") or "")


def pod_limits(pod: dict) -> Dict[str, object]:
    limits: Dict[str, object] = {}
    for c in pod.get("spec", {}).get("containers", []) or []:
        limits.update((c.get("resources", {}) or {}).get("limits", {}) or {})
    return limits


def ungate_pod(pod: dict) -> dict:
    """Remove our scheduling gate (reference: unGatePod,
    instaslice_controller.go:426-433). Returns a modified copy."""
    from instaslice_amd import GATE_NAME

    pod = copy.deepcopy(pod)
    gates = pod.get("spec", {}).get("schedulingGates", []) or []
    pod["spec"]["schedulingGates"] = [g for g in gates if g.get("name") != GATE_NAME]
    return pod


def remove_finalizer(pod: dict) -> dict:
    from instaslice_amd import FINALIZER_NAME

    pod = copy.deepcopy(pod)
    fins = pod.get("metadata", {}).get("finalizers", []) or []
    pod["metadata"]["finalizers"] = [f for f in fins if f != FINALIZER_NAME]
    return pod
