"""MI355X partition model: compute/memory modes, profile catalog, placements.

This is the MI355X-native analog of the reference's MIG profile machinery
(reference: internal/controller/instaslice_daemonset.go:67-93 MigProfile,
:751-783 canonical naming, :588-664 profile discovery). The semantic
difference: NVIDIA MIG carves one slice at a time next
to running slices; AMD partitioning is a *whole-GPU mode*:

  compute mode:  SPX (1 partition of 8 XCDs), DPX (2x4), TPX (where supported),
                 QPX (4x2), CPX (8x1) -- amdsmi_compute_partition_type_t
  memory mode:   NPS1 / NPS2 / NPS4 -- amdsmi_memory_partition_type_t

A *profile* here is (compute mode, XCDs per partition, GB per partition), with
a preferred memory mode. A *placement* on a GPU is a partition ordinal under
the GPU's current mode; occupancy is tracked as an 8-bit XCD bitmap exactly
like the reference's 8-slot MIG bitmap (instaslice_controller.go:306-343),
except validity is governed by the mode, not by per-profile {start,size}
tables.

The catalog is *discovery-driven* on real hardware
(amdsmi_get_gpu_accelerator_partition_profile_config, amdsmi.h:5950); the
static MI355X table below is both the FakeAmdSmi model and the fallback
when the driver cannot enumerate profiles (e.g. in a VM guest).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional, Tuple


class ComputeMode(str, Enum):
    """Compute partition modes (amdsmi_compute_partition_type_t, amdsmi.h:432)."""

    SPX = "SPX"  # 1 partition, all 8 XCDs
    DPX = "DPX"  # 2 partitions, 4 XCDs each
    TPX = "TPX"  # 3 partitions (MI300A only; not valid on MI355X)
    QPX = "QPX"  # 4 partitions, 2 XCDs each
    CPX = "CPX"  # 8 partitions, 1 XCD each

    @property
    def num_partitions(self) -> int:
        return {"SPX": 1, "DPX": 2, "TPX": 3, "QPX": 4, "CPX": 8}[self.value]


class MemoryMode(str, Enum):
    """Memory (NUMA-per-socket) partition modes (amdsmi.h:454)."""

    NPS1 = "NPS1"  # all HBM stacks interleaved: one memory domain
    NPS2 = "NPS2"  # two domains
    NPS4 = "NPS4"  # four domains: best locality for QPX/CPX partitions

    @property
    def num_domains(self) -> int:
        return {"NPS1": 1, "NPS2": 2, "NPS4": 4}[self.value]


# MI355X chip constants (MI355X_MICROARCH.md: 8 XCDs, 288 GB HBM3E).
MI355X_XCD_COUNT = 8
MI355X_HBM_GB = 288
MI355X_NAME = "AMD Instinct MI355X"

# Memory modes each compute mode may combine with. NPS4 needs >= 4 compute
# partitions so each partition maps onto whole HBM quadrants; NPS2 likewise
# needs >= 2. (Model mirrors amd-smi's advertised memory_caps per profile.)
VALID_MEMORY_MODES: Dict[ComputeMode, Tuple[MemoryMode, ...]] = {
    ComputeMode.SPX: (MemoryMode.NPS1,),
    ComputeMode.DPX: (MemoryMode.NPS1, MemoryMode.NPS2),
    ComputeMode.QPX: (MemoryMode.NPS1, MemoryMode.NPS4),
    ComputeMode.CPX: (MemoryMode.NPS1, MemoryMode.NPS4),
}

_PROFILE_RE = re.compile(r"^(spx|dpx|tpx|qpx|cpx)-(\d+)x(\d+)$")


@dataclass(frozen=True)
class PartitionProfile:
    """A requestable partition shape, canonically named `<mode>-<xcds>x<gb>`.

    Examples on a 288 GB MI355X: cpx-1x36, qpx-2x72, dpx-4x144, spx-8x288.
    The reference's analog is MigProfile with its `1g.5gb` naming
    (instaslice_daemonset.go:774-783); GB here is exact (total/partitions),
    not the reference's 1/8-rounding (:763-771), because AMD partitions
    split HBM evenly by construction.
    """

    compute: ComputeMode
    xcds: int  # XCDs per partition
    memory_gb: int  # HBM GB per partition
    preferred_memory: MemoryMode = MemoryMode.NPS1
    # amd-smi profile_index for amdsmi_set_gpu_accelerator_partition_profile,
    # when discovered from hardware; None for static-catalog entries.
    profile_index: Optional[int] = None

    @property
    def name(self) -> str:
        return f"{self.compute.value.lower()}-{self.xcds}x{self.memory_gb}"

    @property
    def partitions_per_gpu(self) -> int:
        return self.compute.num_partitions

    @property
    def resource_name(self) -> str:
        """Extended-resource key a pod requests (reference analog:
        `nvidia.com/mig-1g.5gb`, samples/test-pod.yaml:13)."""
        from instaslice_amd import RESOURCE_PREFIX

        return RESOURCE_PREFIX + self.name


@dataclass
class ProfileCatalog:
    """Discovered (or modeled) set of profiles for one GPU model.

    Reference analog: the `Migplacement` list persisted into the Instaslice
    CR at discovery time (instaslice_daemonset.go:642-658).
    """

    gpu_model: str
    total_memory_gb: int
    xcd_count: int
    profiles: List[PartitionProfile] = field(default_factory=list)

    def by_name(self, name: str) -> Optional[PartitionProfile]:
        for p in self.profiles:
            if p.name == name:
                return p
        return None

    def to_dict(self) -> dict:
        return {
            "gpu_model": self.gpu_model,
            "total_memory_gb": self.total_memory_gb,
            "xcd_count": self.xcd_count,
            "profiles": [
                {
                    "name": p.name,
                    "compute": p.compute.value,
                    "xcds": p.xcds,
                    "memory_gb": p.memory_gb,
                    "preferred_memory": p.preferred_memory.value,
                    "profile_index": p.profile_index,
                }
                for p in self.profiles
            ],
        }

    @classmethod
    def from_dict(cls, d: dict) -> "ProfileCatalog":
        cat = cls(
            gpu_model=d["gpu_model"],
            total_memory_gb=d["total_memory_gb"],
            xcd_count=d["xcd_count"],
        )
        for pd in d.get("profiles", []):
            cat.profiles.append(
                PartitionProfile(
                    compute=ComputeMode(pd["compute"]),
                    xcds=pd["xcds"],
                    memory_gb=pd["memory_gb"],
                    preferred_memory=MemoryMode(pd["preferred_memory"]),
                    profile_index=pd.get("profile_index"),
                )
            )
        return cat


def mi355x_catalog() -> ProfileCatalog:
    """Static MI355X catalog: the FakeAmdSmi model and the VM-guest fallback.

    QPX/CPX prefer NPS4 (each partition gets local HBM quadrants -> lower
    latency, full per-XCD bandwidth); SPX/DPX prefer NPS1.
    """
    cat = ProfileCatalog(
        gpu_model=MI355X_NAME, total_memory_gb=MI355X_HBM_GB, xcd_count=MI355X_XCD_COUNT
    )
    for mode, pref in (
        (ComputeMode.SPX, MemoryMode.NPS1),
        (ComputeMode.DPX, MemoryMode.NPS1),
        (ComputeMode.QPX, MemoryMode.NPS4),
        (ComputeMode.CPX, MemoryMode.NPS4),
    ):
        n = mode.num_partitions
        cat.profiles.append(
            PartitionProfile(
                compute=mode,
                xcds=MI355X_XCD_COUNT // n,
                memory_gb=MI355X_HBM_GB // n,
                preferred_memory=pref,
            )
        )
    return cat


def catalog_from_amdsmi_profiles(
    gpu_model: str, total_memory_gb: int, raw_profiles: List[dict]
) -> ProfileCatalog:
    """Build a catalog from amd-smi accelerator-partition profile dicts.

    `raw_profiles` entries carry {"profile_type": "SPX"|..., "num_partitions",
    "profile_index", "memory_caps": ["NPS1", ...]} as produced by the
    partitiond shim (smi/native.py) or FakeAmdSmi.
    """
    cat = ProfileCatalog(
        gpu_model=gpu_model,
        total_memory_gb=total_memory_gb,
        xcd_count=MI355X_XCD_COUNT,
    )
    for rp in raw_profiles:
        try:
            mode = ComputeMode(rp["profile_type"])
        except ValueError:
            continue  # unknown/future mode: skip rather than fail discovery
        n = int(rp["num_partitions"]) or mode.num_partitions
        caps = [MemoryMode(m) for m in rp.get("memory_caps", []) if m in MemoryMode._value2member_map_]
        pref = MemoryMode.NPS4 if MemoryMode.NPS4 in caps and n >= 4 else MemoryMode.NPS1
        cat.profiles.append(
            PartitionProfile(
                compute=mode,
                xcds=max(1, cat.xcd_count // n),
                memory_gb=max(1, total_memory_gb // n),
                preferred_memory=pref,
                profile_index=rp.get("profile_index"),
            )
        )
    if not cat.profiles:  # driver gave nothing usable: fall back to the model
        return mi355x_catalog()
    return cat


def parse_profile_name(name: str) -> Tuple[ComputeMode, int, int]:
    """Parse `cpx-1x36` -> (ComputeMode.CPX, 1, 36). Raises ValueError.

    Reference analog: extractProfileName's regex `(\\d+g\\.\\d+gb)`
    (instaslice_controller.go:265-280).
    """
    m = _PROFILE_RE.match(name)
    if not m:
        raise ValueError(f"not a partition profile name: {name!r}")
    return ComputeMode(m.group(1).upper()), int(m.group(2)), int(m.group(3))


def extract_profile_from_limits(limits: Dict[str, object]) -> Optional[str]:
    """Find the partition-profile request in a pod's resource limits.

    Returns the profile name (e.g. "cpx-1x36") or None. Exactly one partition
    resource may be requested per pod (mirrors the reference's single-profile
    assumption, instaslice_controller.go:154).
    """
    from instaslice_amd import RESOURCE_PREFIX

    found = []
    for key in limits:
        if key.startswith(RESOURCE_PREFIX):
            name = key[len(RESOURCE_PREFIX):]
            if _PROFILE_RE.match(name):
                found.append(name)
    if not found:
        return None
    if len(found) > 1:
        raise ValueError(f"pod requests multiple partition profiles: {found}")
    return found[0]


def xcd_mask(ordinal: int, xcds_per_partition: int) -> int:
    """XCD occupancy bitmap for partition `ordinal` in a mode with
    `xcds_per_partition` XCDs per partition. Partition k of a CPX GPU
    occupies XCD k; partition k of QPX occupies XCDs 2k..2k+1; etc."""
    start = ordinal * xcds_per_partition
    return ((1 << xcds_per_partition) - 1) << start
