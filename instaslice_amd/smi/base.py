"""L3 device-layer interface: the verbs the node agent needs from amd-smi.

This is the MI355X-native replacement for the reference's NVML surface
(go-nvml cgo bindings, instaslice_daemonset.go:29,40,62-65; call-site table
in SURVEY.md §2.2). Two implementations:

  - smi/native.py: C++ "partitiond" shim linking /opt/rocm/lib/libamd_smi.so
    (first-party native code; the reference's native layer was NVML reached
    from Go)
  - smi/fake.py:  FakeAmdSmi modeling an N x MI355X node, the analog of the
    reference's dgxa100 NVML mock (instaslice_daemonset_test.go:39)

Design rule carried over from the north star: enumeration is performed ONCE
and cached by the caller (the reference re-runs nvml.Init on every reconcile,
instaslice_daemonset.go:112 — the hot spot SURVEY.md §3.2 calls out).
"""

from __future__ import annotations

import abc
from dataclasses import dataclass, field
from typing import Dict, List


class SmiError(Exception):
    """Base for device-layer failures. Unlike the reference (which logs and
    swallows NVML errors, instaslice_daemonset.go:173-189), these propagate so
    the reconciler can fail the allocation and roll back."""


class SmiBusy(SmiError):
    """Partition mode change refused: device not idle (EBUSY analog)."""


class SmiNotSupported(SmiError):
    """Operation unsupported on this device/driver (e.g. VM guest)."""


class SmiPermission(SmiError):
    """Operation requires elevated privileges."""


@dataclass
class PartitionDevice:
    """One schedulable sub-device of a physical GPU (a CPX/QPX/DPX partition,
    or the whole GPU in SPX)."""

    uuid: str          # device UUID the workload sees (ROCR_VISIBLE_DEVICES)
    ordinal: int       # partition ordinal within the physical GPU (0-based)
    device_index: int  # node-wide HIP device index (for in-process payloads)
    memory_gb: int


@dataclass
class PhysicalGpu:
    """One physical GPU package with its current partition state."""

    uuid: str          # stable physical-GPU UUID (discovered at boot, SPX view)
    index: int         # physical index on the node
    model: str
    memory_gb: int
    compute_mode: str  # "SPX" | "DPX" | "QPX" | "CPX"
    memory_mode: str   # "NPS1" | "NPS2" | "NPS4"
    partitions: List[PartitionDevice] = field(default_factory=list)


class AmdSmi(abc.ABC):
    """The 13-verb device interface."""

    @abc.abstractmethod
    def init(self) -> None:
        """Attach to the driver (amdsmi_init). Idempotent."""

    @abc.abstractmethod
    def shutdown(self) -> None:
        """Detach (amdsmi_shut_down). Idempotent."""

    @abc.abstractmethod
    def list_gpus(self) -> List[PhysicalGpu]:
        """Enumerate physical GPUs with current modes and partition devices.
        (amdsmi_get_socket_handles + amdsmi_get_processor_handles +
        amdsmi_get_gpu_device_uuid/_asic_info/_memory_total, grouped by
        physical package.) Callers cache the result; re-call only after a
        mode change re-enumerates partitions."""

    @abc.abstractmethod
    def get_compute_partition(self, gpu_uuid: str) -> str:
        """amdsmi_get_gpu_compute_partition (amdsmi.h:5768)."""

    @abc.abstractmethod
    def set_compute_partition(self, gpu_uuid: str, mode: str) -> None:
        """amdsmi_set_gpu_compute_partition (amdsmi.h:5799). Whole-GPU; the
        device must be idle — raises SmiBusy otherwise."""

    @abc.abstractmethod
    def get_memory_partition(self, gpu_uuid: str) -> str:
        """amdsmi_get_gpu_memory_partition (amdsmi.h:5844)."""

    @abc.abstractmethod
    def set_memory_partition(self, gpu_uuid: str, mode: str) -> None:
        """amdsmi_set_gpu_memory_partition (amdsmi.h:5876). NOTE: on bare
        metal this requires an amdgpu driver reload; treat memory mode as
        sticky per GPU."""

    @abc.abstractmethod
    def get_profile_config(self, gpu_uuid: str) -> List[dict]:
        """Accelerator-partition profile catalog
        (amdsmi_get_gpu_accelerator_partition_profile_config, amdsmi.h:5950).
        Returns [{"profile_type", "num_partitions", "profile_index",
        "memory_caps": [...]}]."""

    @abc.abstractmethod
    def get_metrics(self, gpu_uuid: str) -> Dict[str, float]:
        """Activity/VRAM counters for observability (captured around every
        reconfigure, per the north star)."""

    def get_topology(self) -> Dict[str, Dict[str, int]]:
        """Physical-GPU link topology: {src_uuid: {dst_uuid: hops}} for
        XGMI-class links (amdsmi_topo_get_link_type, amdsmi.h:5635).
        Placement uses it for hop-aware gang scoring. Default: empty
        (topology unknown — scoring degrades to same-GPU affinity only)."""
        return {}

    def set_accelerator_profile(self, gpu_uuid: str, profile_index: int) -> None:
        """Set the partition layout by catalog profile index
        (amdsmi_set_gpu_accelerator_partition_profile, amdsmi.h:5994) — the
        documented forward path; the amd-smi CLI routes `set -C` through it.
        Default: resolve the index to its mode via get_profile_config and
        delegate to set_compute_partition (exact semantics for the fake);
        NativeAmdSmi overrides with the direct amdsmi call."""
        for p in self.get_profile_config(gpu_uuid):
            if p["profile_index"] == profile_index:
                self.set_compute_partition(gpu_uuid, p["profile_type"])
                return
        raise SmiError(
            f"gpu {gpu_uuid}: no profile with index {profile_index} in catalog"
        )

    # -- convenience ------------------------------------------------------

    def find_gpu(self, gpu_uuid: str) -> PhysicalGpu:
        for g in self.list_gpus():
            if g.uuid == gpu_uuid:
                return g
        raise SmiError(f"gpu {gpu_uuid} not found")
