"""FakeAmdSmi: an in-memory model of an N x MI355X node.

The analog of the reference's dgxa100 NVML mock server
(instaslice_daemonset_test.go:25,39) — but richer, per SURVEY.md §4's
implication list: it models

  - the full partition matrix (SPX/DPX/QPX/CPX x NPS1/NPS4) with validity
    rules (partition/profiles.VALID_MEMORY_MODES)
  - EBUSY when a mode change is attempted on a busy GPU
  - configurable mode-change latency (compute flip ~ms, memory flip ~slow,
    modeling the driver-reload requirement of amdsmi_set_gpu_memory_partition)
  - CPX re-enumeration: after a mode set, one physical GPU exposes
    num_partitions sub-devices with fresh deterministic UUIDs and re-packed
    node-wide HIP device indices
  - injectable failures for fault-injection tests (the reference has none,
    SURVEY.md §5)
"""

from __future__ import annotations

import threading
import time
import uuid as uuidlib
from typing import Callable, Dict, List, Optional

from instaslice_amd.partition.profiles import (
    MI355X_HBM_GB,
    MI355X_NAME,
    ComputeMode,
    MemoryMode,
    VALID_MEMORY_MODES,
)
from instaslice_amd.smi.base import (
    AmdSmi,
    PartitionDevice,
    PhysicalGpu,
    SmiBusy,
    SmiError,
    SmiNotSupported,
)

_NAMESPACE = uuidlib.UUID("6ba7b810-9dad-11d1-80b4-00c04fd430c8")


def _det_uuid(*parts: object) -> str:
    """Deterministic UUID so re-enumeration after the same mode sequence is
    stable across processes (needed by multi-rank tests)."""
    return str(uuidlib.uuid5(_NAMESPACE, ":".join(str(p) for p in parts)))


class _FakeGpu:
    def __init__(self, node: str, index: int, memory_gb: int):
        self.node = node
        self.index = index
        self.memory_gb = memory_gb
        self.uuid = _det_uuid(node, "gpu", index)
        self.compute_mode = ComputeMode.SPX
        self.memory_mode = MemoryMode.NPS1
        # uuids of partitions with running workloads (set by tests / agent)
        self.busy_partitions: set = set()
        self.mode_generation = 0  # bumps on every mode change


class FakeAmdSmi(AmdSmi):
    def __init__(
        self,
        num_gpus: int = 8,
        node_name: str = "node-0",
        memory_gb: int = MI355X_HBM_GB,
        compute_set_latency_s: float = 0.0,
        memory_set_latency_s: float = 0.0,
        xgmi_cliques: Optional[list] = None,
    ):
        self._lock = threading.RLock()
        self._gpus = [_FakeGpu(node_name, i, memory_gb) for i in range(num_gpus)]
        # xGMI topology model: GPUs in one clique are 1-hop neighbors. A
        # real MI355X node is one fully-connected clique of 8 (7 p2p links
        # per GPU); tests inject partial meshes (e.g. [[0,1,2,3],[4,5,6,7]])
        # to exercise hop-aware gang scoring.
        self.xgmi_cliques = (
            xgmi_cliques if xgmi_cliques is not None
            else [list(range(num_gpus))]
        )
        self._initialized = False
        self.compute_set_latency_s = compute_set_latency_s
        self.memory_set_latency_s = memory_set_latency_s
        # fault injection: callable(verb, gpu_uuid) may raise
        self.fault_hook: Optional[Callable[[str, str], None]] = None
        # counters for assertions (e.g. "enumeration happened once")
        self.call_counts: Dict[str, int] = {}
        # flip-batching observability: how many mode sets ran CONCURRENTLY
        # at peak (asserted by the agent's flip-batching test)
        self.max_concurrent_sets = 0
        self._conc = 0
        self._conc_lock = threading.Lock()

    # -- internals ----------------------------------------------------------

    def _count(self, verb: str, gpu_uuid: str = "") -> None:
        self.call_counts[verb] = self.call_counts.get(verb, 0) + 1
        if self.fault_hook:
            self.fault_hook(verb, gpu_uuid)

    def _gpu(self, gpu_uuid: str) -> _FakeGpu:
        for g in self._gpus:
            if g.uuid == gpu_uuid:
                return g
        raise SmiError(f"gpu {gpu_uuid} not found")

    def _require_init(self) -> None:
        if not self._initialized:
            raise SmiError("amdsmi not initialized")

    # -- test hooks ----------------------------------------------------------

    def mark_busy(self, gpu_uuid: str, partition_uuid: str, busy: bool = True) -> None:
        """Simulate a workload running in a partition (makes mode changes
        fail with SmiBusy, as the real driver does: amdsmi.h:5781 'Device
        must be idle')."""
        g = self._gpu(gpu_uuid)
        if busy:
            g.busy_partitions.add(partition_uuid)
        else:
            g.busy_partitions.discard(partition_uuid)

    # -- AmdSmi interface ------------------------------------------------

    def init(self) -> None:
        self._count("init")
        self._initialized = True

    def shutdown(self) -> None:
        self._count("shutdown")
        self._initialized = False

    def list_gpus(self) -> List[PhysicalGpu]:
        self._require_init()
        self._count("list_gpus")
        with self._lock:
            out: List[PhysicalGpu] = []
            device_index = 0  # node-wide HIP ordinal packing, like ROCm does
            for g in self._gpus:
                n = g.compute_mode.num_partitions
                parts = []
                for k in range(n):
                    parts.append(
                        PartitionDevice(
                            uuid=_det_uuid(g.uuid, g.mode_generation, g.compute_mode.value, k),
                            ordinal=k,
                            device_index=device_index,
                            memory_gb=g.memory_gb // n,
                        )
                    )
                    device_index += 1
                out.append(
                    PhysicalGpu(
                        uuid=g.uuid,
                        index=g.index,
                        model=MI355X_NAME,
                        memory_gb=g.memory_gb,
                        compute_mode=g.compute_mode.value,
                        memory_mode=g.memory_mode.value,
                        partitions=parts,
                    )
                )
            return out

    def get_compute_partition(self, gpu_uuid: str) -> str:
        self._require_init()
        self._count("get_compute_partition", gpu_uuid)
        return self._gpu(gpu_uuid).compute_mode.value

    def set_compute_partition(self, gpu_uuid: str, mode: str) -> None:
        self._require_init()
        self._count("set_compute_partition", gpu_uuid)
        try:
            new_mode = ComputeMode(mode)
        except ValueError:
            raise SmiError(f"invalid compute mode {mode!r}")
        if new_mode is ComputeMode.TPX:
            raise SmiNotSupported("TPX is not supported on MI355X")
        with self._lock:
            g = self._gpu(gpu_uuid)
            if g.busy_partitions:
                raise SmiBusy(
                    f"gpu {gpu_uuid}: {len(g.busy_partitions)} partitions busy"
                )
            if new_mode is g.compute_mode:
                return
        # latency OUTSIDE the node lock: on real hardware mode sets on
        # DIFFERENT GPUs proceed independently, and the agent's flip
        # batching relies on that (peak concurrency is recorded for the
        # batching test)
        with self._conc_lock:
            self._conc += 1
            self.max_concurrent_sets = max(self.max_concurrent_sets, self._conc)
        try:
            if self.compute_set_latency_s:
                time.sleep(self.compute_set_latency_s)
        finally:
            with self._conc_lock:
                self._conc -= 1
        with self._lock:
            g = self._gpu(gpu_uuid)
            if g.busy_partitions:
                raise SmiBusy(f"gpu {gpu_uuid}: busy")
            if new_mode is g.compute_mode:
                return
            g.compute_mode = new_mode
            g.mode_generation += 1
            # mode combination validity: keep memory mode legal
            if g.memory_mode not in VALID_MEMORY_MODES[new_mode]:
                g.memory_mode = MemoryMode.NPS1

    def get_memory_partition(self, gpu_uuid: str) -> str:
        self._require_init()
        self._count("get_memory_partition", gpu_uuid)
        return self._gpu(gpu_uuid).memory_mode.value

    def set_memory_partition(self, gpu_uuid: str, mode: str) -> None:
        self._require_init()
        self._count("set_memory_partition", gpu_uuid)
        try:
            new_mode = MemoryMode(mode)
        except ValueError:
            raise SmiError(f"invalid memory mode {mode!r}")
        with self._lock:
            g = self._gpu(gpu_uuid)
            if g.busy_partitions:
                raise SmiBusy(f"gpu {gpu_uuid}: busy")
            if new_mode not in VALID_MEMORY_MODES[g.compute_mode]:
                raise SmiNotSupported(
                    f"{new_mode.value} invalid under {g.compute_mode.value}"
                )
            if new_mode is g.memory_mode:
                return
            if self.memory_set_latency_s:
                time.sleep(self.memory_set_latency_s)
            g.memory_mode = new_mode
            g.mode_generation += 1

    def get_profile_config(self, gpu_uuid: str) -> List[dict]:
        self._require_init()
        self._count("get_profile_config", gpu_uuid)
        self._gpu(gpu_uuid)
        out = []
        for idx, mode in enumerate(
            (ComputeMode.SPX, ComputeMode.DPX, ComputeMode.QPX, ComputeMode.CPX)
        ):
            out.append(
                {
                    "profile_type": mode.value,
                    "num_partitions": mode.num_partitions,
                    "profile_index": idx,
                    "memory_caps": [m.value for m in VALID_MEMORY_MODES[mode]],
                }
            )
        return out

    def get_topology(self) -> Dict[str, Dict[str, int]]:
        self._require_init()
        self._count("get_topology")
        topo: Dict[str, Dict[str, int]] = {}
        for clique in self.xgmi_cliques:
            for i in clique:
                for j in clique:
                    if i == j:
                        continue
                    topo.setdefault(self._gpus[i].uuid, {})[
                        self._gpus[j].uuid] = 1
        return topo

    def get_metrics(self, gpu_uuid: str) -> Dict[str, float]:
        self._require_init()
        self._count("get_metrics", gpu_uuid)
        g = self._gpu(gpu_uuid)
        return {
            "gfx_activity_pct": 37.0 if g.busy_partitions else 0.0,
            "vram_used_mb": 1024.0 * len(g.busy_partitions),
            "mode_generation": float(g.mode_generation),
        }
