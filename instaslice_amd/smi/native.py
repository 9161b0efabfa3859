"""NativeAmdSmi: the AmdSmi interface over the C++ partitiond shim.

Grouping policy lives here (unit-testable): amd-smi enumerates *processors*
(one per partition — a CPX GPU shows up 8 times), while the control plane
reasons about *physical GPUs*. Processors are grouped into packages by
asic_serial (partitions of one package share the serial); the stable
physical-GPU UUID is the UUID of the partition_id==0 processor recorded at
first enumeration, so it survives mode flips even though partition UUIDs
change (SURVEY.md §7.3 're-enumeration after CPX').

Fallbacks for VM guests where serial/partition_id may be unpopulated:
processor-count identity mapping (1 processor == 1 GPU).
"""

from __future__ import annotations

import threading
from typing import Dict, List

from instaslice_amd.smi.base import (
    AmdSmi,
    PartitionDevice,
    PhysicalGpu,
    SmiBusy,
    SmiError,
    SmiNotSupported,
    SmiPermission,
)
from instaslice_amd.utils import get_logger

try:
    from instaslice_amd.smi import _partitiond as _pd

    NATIVE_AVAILABLE = True
except ImportError as _e:  # extension not built
    _pd = None
    NATIVE_AVAILABLE = False
    _IMPORT_ERROR = _e


def _translate(exc: Exception) -> Exception:
    """Map amdsmi status codes onto the typed SmiError hierarchy."""
    args = getattr(exc, "args", ())
    status = None
    if len(args) >= 1 and isinstance(args[0], tuple) and len(args[0]) == 2:
        msg, status = args[0]
    elif len(args) == 2 and isinstance(args[1], int):
        msg, status = args
    else:
        msg = str(args[0]) if args else str(exc)
    if status == _pd.STATUS_BUSY:
        return SmiBusy(msg)
    if status == _pd.STATUS_NO_PERM:
        return SmiPermission(msg)
    if status in (_pd.STATUS_NOT_SUPPORTED, _pd.STATUS_SETTING_UNAVAILABLE):
        return SmiNotSupported(msg)
    return SmiError(msg)


class NativeAmdSmi(AmdSmi):
    def __init__(self) -> None:
        if not NATIVE_AVAILABLE:
            raise SmiError(
                f"_partitiond extension not built (run build_native.py): {_IMPORT_ERROR}"
            )
        self._d = _pd.Partitiond()
        self._lock = threading.RLock()
        self.log = get_logger("smi.native")
        # serial -> stable physical uuid, recorded at first enumeration
        self._serial_to_uuid: Dict[str, str] = {}
        self._boot_uuids: List[str] = []  # identity-mapping fallback
        # physical uuid -> processor index to address for mode sets
        self._uuid_to_proc_index: Dict[str, int] = {}

    # -- lifecycle ----------------------------------------------------------

    def init(self) -> None:
        try:
            self._d.init()
        except _pd.SmiNativeError as e:
            raise _translate(e) from None

    def shutdown(self) -> None:
        try:
            self._d.shutdown()
        except _pd.SmiNativeError as e:
            raise _translate(e) from None

    # -- enumeration ----------------------------------------------------------

    def _group(self, procs: list) -> Dict[str, list]:
        """Group processor records into physical packages. Returns
        {group_key: [proc, ...]} with procs ordered by partition_id."""
        serials = [p.asic_serial for p in procs]
        have_serials = all(serials) and (
            len(set(serials)) < len(serials) or len(procs) == len(set(serials))
        )
        groups: Dict[str, list] = {}
        if have_serials:
            for p in procs:
                groups.setdefault(p.asic_serial, []).append(p)
        else:
            # VM fallback: serials absent -> each processor its own package
            for p in procs:
                groups.setdefault(f"proc-{p.index}", []).append(p)
        for g in groups.values():
            g.sort(
                key=lambda p: p.partition_id if p.partition_id != 0xFFFFFFFF else p.index
            )
        return groups

    def list_gpus(self) -> List[PhysicalGpu]:
        with self._lock:
            try:
                procs = self._d.enumerate()
            except _pd.SmiNativeError as e:
                raise _translate(e) from None
            if not procs:
                return []
            groups = self._group(procs)
            out: List[PhysicalGpu] = []
            for idx, (key, members) in enumerate(
                sorted(groups.items(), key=lambda kv: kv[1][0].index)
            ):
                head = members[0]
                # stable physical uuid: first-seen partition-0 uuid per serial
                if key not in self._serial_to_uuid:
                    self._serial_to_uuid[key] = head.uuid
                phys_uuid = self._serial_to_uuid[key]
                self._uuid_to_proc_index[phys_uuid] = head.index
                mode = head.compute_partition or "SPX"
                mem_mode = head.memory_partition or "NPS1"
                parts = [
                    PartitionDevice(
                        uuid=p.uuid,
                        ordinal=(
                            p.partition_id if p.partition_id != 0xFFFFFFFF else k
                        ),
                        device_index=p.index,
                        memory_gb=int(p.vram_total_mb // 1024),
                    )
                    for k, p in enumerate(members)
                ]
                total_gb = sum(pt.memory_gb for pt in parts)
                out.append(
                    PhysicalGpu(
                        uuid=phys_uuid,
                        index=idx,
                        model=head.asic_name or "AMD GPU",
                        memory_gb=total_gb,
                        compute_mode=mode,
                        memory_mode=mem_mode,
                        partitions=parts,
                    )
                )
            return out

    def _proc_index(self, gpu_uuid: str) -> int:
        if gpu_uuid not in self._uuid_to_proc_index:
            self.list_gpus()
        if gpu_uuid not in self._uuid_to_proc_index:
            raise SmiError(f"gpu {gpu_uuid} not found")
        return self._uuid_to_proc_index[gpu_uuid]

    # -- partition verbs -----------------------------------------------------

    def get_compute_partition(self, gpu_uuid: str) -> str:
        try:
            return self._d.get_compute_partition(self._proc_index(gpu_uuid))
        except _pd.SmiNativeError as e:
            raise _translate(e) from None

    def set_compute_partition(self, gpu_uuid: str, mode: str) -> None:
        with self._lock:
            idx = self._proc_index(gpu_uuid)
            try:
                self._d.set_compute_partition(idx, mode)
            except _pd.SmiNativeError as e:
                first = _translate(e)
                if isinstance(first, (SmiBusy, SmiPermission)):
                    raise first from None
                # The type-based set (amdsmi_set_gpu_compute_partition) is
                # refused on some platforms where the PROFILE-INDEX set
                # still works — the amd-smi CLI routes `set -C` through
                # amdsmi_set_gpu_accelerator_partition_profile and succeeds
                # where the type set returns UNKNOWN_ERROR (probe evidence:
                # profiles/partition_flip_probe_r01.txt line 39-41). Fall
                # back to the index API before giving up.
                try:
                    pidx = self._profile_index_for(idx, mode)
                except SmiError:
                    raise first from None
                if pidx is None:
                    raise SmiNotSupported(
                        f"gpu {gpu_uuid}: mode {mode} not in profile catalog "
                        f"(type-set also failed: {first})"
                    ) from None
                self.log.info(
                    "gpu %s: type-based set %s failed (%s); retrying via "
                    "profile index %d", gpu_uuid[:8], mode, first, pidx,
                )
                try:
                    self._d.set_accelerator_profile(idx, pidx)
                except _pd.SmiNativeError as e2:
                    second = _translate(e2)
                    raise type(second)(
                        f"both partition set APIs failed for {mode}: "
                        f"type-based [{first}]; profile-index {pidx} "
                        f"[{second}]"
                    ) from None
            # processor population changed: stale index cache
            self._uuid_to_proc_index.pop(gpu_uuid, None)

    def _profile_index_for(self, proc_index: int, mode: str):
        """Catalog profile_index for a compute mode, or None if absent."""
        try:
            raw = self._d.get_profile_config(proc_index)
        except _pd.SmiNativeError as e:
            raise _translate(e) from None
        for p in raw:
            if p.profile_type == mode:
                return p.profile_index
        return None

    def set_accelerator_profile(self, gpu_uuid: str, profile_index: int) -> None:
        with self._lock:
            try:
                self._d.set_accelerator_profile(
                    self._proc_index(gpu_uuid), profile_index
                )
            except _pd.SmiNativeError as e:
                raise _translate(e) from None
            self._uuid_to_proc_index.pop(gpu_uuid, None)

    def get_memory_partition(self, gpu_uuid: str) -> str:
        try:
            return self._d.get_memory_partition(self._proc_index(gpu_uuid))
        except _pd.SmiNativeError as e:
            raise _translate(e) from None

    def set_memory_partition(self, gpu_uuid: str, mode: str) -> None:
        with self._lock:
            try:
                self._d.set_memory_partition(self._proc_index(gpu_uuid), mode)
            except _pd.SmiNativeError as e:
                raise _translate(e) from None
            self._uuid_to_proc_index.pop(gpu_uuid, None)

    def get_profile_config(self, gpu_uuid: str) -> List[dict]:
        try:
            raw = self._d.get_profile_config(self._proc_index(gpu_uuid))
        except _pd.SmiNativeError as e:
            raise _translate(e) from None
        return [
            {
                "profile_type": p.profile_type,
                "num_partitions": p.num_partitions,
                "profile_index": p.profile_index,
                "memory_caps": list(p.memory_caps),
            }
            for p in raw
        ]

    def get_topology(self) -> Dict[str, Dict[str, int]]:
        """xGMI topology between physical packages: {src_uuid: {dst_uuid:
        hops}}. Processor-level links are aggregated to the package level
        (in CPX one package enumerates as 8 processors; intra-package links
        are INTERNAL and skipped)."""
        with self._lock:
            try:
                procs = self._d.enumerate()
                links = self._d.get_link_topology()
            except _pd.SmiNativeError as e:
                raise _translate(e) from None
            groups = self._group(procs)
            proc_to_phys: Dict[int, str] = {}
            for key, members in groups.items():
                head = members[0]
                phys = self._serial_to_uuid.get(key, head.uuid)
                for p in members:
                    proc_to_phys[p.index] = phys
            topo: Dict[str, Dict[str, int]] = {}
            for li in links:
                if li.type not in ("XGMI",):
                    continue
                s = proc_to_phys.get(li.src)
                d = proc_to_phys.get(li.dst)
                if s is None or d is None or s == d:
                    continue
                cur = topo.setdefault(s, {})
                hops = int(li.hops)
                if d not in cur or hops < cur[d]:
                    cur[d] = hops
            return topo

    def get_metrics(self, gpu_uuid: str) -> Dict[str, float]:
        try:
            m = self._d.get_metrics(self._proc_index(gpu_uuid))
        except _pd.SmiNativeError as e:
            raise _translate(e) from None
        return {
            "gfx_activity_pct": m.gfx_activity_pct,
            "umc_activity_pct": m.umc_activity_pct,
            "vram_used_mb": m.vram_used_mb,
            "socket_power_w": m.socket_power_w,
        }
