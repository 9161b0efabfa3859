#!/usr/bin/env python3
"""BASELINE.json measurement configs as runnable scenarios.

  plumbing   config 1: fake-GPU inventory, 1 pod, reconcile-only latency
  cpx8       config 2: one GPU in CPX, 8 sleep pods, one per partition
  mixed100   config 3: 8 GPUs, mixed CPX/QPX profiles, 100 pods, bin-pack
  vllm       config 4: vLLM-shaped pod (samples/vllm_dep.yaml analog) on a
             2-XCD partition
  churn      config 5: N create/delete cycles across 8 GPUs with a live pool
             of pods; reports reconfig count + fragmentation

All scenarios run against FakeAmdSmi by default so they execute anywhere
(bench.py is the real-GPU entry point; scenarios quantify policy behavior).
Run: python -m benchmarks.scenarios [--scenario all] [--pods N]
     [--policy P] [--reconfig-latency S]   (output is JSON lines)
"""

from __future__ import annotations

import argparse
import json
import random
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from instaslice_amd.api.types import AllocationStatus  # noqa: E402
from instaslice_amd.controller.reconciler import INSTASLICE_NS  # noqa: E402
from instaslice_amd.runtime.cluster import Cluster  # noqa: E402
from instaslice_amd.smi.fake import FakeAmdSmi  # noqa: E402


def _mk_cluster(num_gpus: int = 8, nodes: int = 1, policy: str = "packed-fit",
                compute_latency_s: float = 0.0) -> Cluster:
    from instaslice_amd.store.native import stored_available

    backend = "native" if stored_available() else "mem"
    c = Cluster(teardown_grace_s=0.0, policy=policy, backend=backend)
    for n in range(nodes):
        c.add_node(
            f"node-{n}",
            FakeAmdSmi(num_gpus=num_gpus, node_name=f"node-{n}",
                       compute_set_latency_s=compute_latency_s),
        )
    return c.start()


def _lat_stats(samples_ms):
    if not samples_ms:
        return {}
    xs = sorted(samples_ms)
    return {
        "p50_ms": round(statistics.median(xs), 3),
        "p90_ms": round(xs[int(0.9 * (len(xs) - 1))], 3),
        "p99_ms": round(xs[int(0.99 * (len(xs) - 1))], 3),
        "max_ms": round(xs[-1], 3),
        "n": len(xs),
    }


def fragmentation(cluster: Cluster, node: str) -> dict:
    """Free-capacity split across mode-pinned GPUs: the AMD fragmentation
    axis is per-GPU-mode, so we report how many XCDs are
    free but pinned inside a non-empty GPU's mode vs free on idle GPUs."""
    cr = cluster.store.get("Instaslice", node, INSTASLICE_NS)
    pinned_free_xcds = 0
    idle_free_xcds = 0
    for gd in cr["spec"]["gpus"].values():
        n_parts = {"SPX": 1, "DPX": 2, "TPX": 3, "QPX": 4, "CPX": 8}[gd["computeMode"]]
        xcds_per = 8 // n_parts
        used = len(gd.get("usedOrdinals", []))
        free = (n_parts - used) * xcds_per
        if used:
            pinned_free_xcds += free
        else:
            idle_free_xcds += free
    total_free = pinned_free_xcds + idle_free_xcds
    return {
        "pinned_free_xcds": pinned_free_xcds,
        "idle_free_xcds": idle_free_xcds,
        "fragmentation_ratio": round(pinned_free_xcds / total_free, 4)
        if total_free else 0.0,
    }


def scenario_plumbing(args) -> dict:
    """1 pod, reconcile-only: the floor of allocation latency."""
    c = _mk_cluster(num_gpus=1)
    lat = []
    try:
        for i in range(args.pods or 50):
            t0 = time.perf_counter()
            c.submit_pod(f"pl-{i}", "spx-8x288")
            c.wait_pod_scheduled(f"pl-{i}")
            lat.append((time.perf_counter() - t0) * 1000)
            c.delete_pod(f"pl-{i}")
            c.wait_pod_gone(f"pl-{i}")
            c.wait_allocations_empty("node-0")
    finally:
        c.stop()
    return {"scenario": "plumbing", "latency": _lat_stats(lat)}


def scenario_cpx8(args) -> dict:
    """8 sleep pods, one per CPX partition of one GPU; 1 reconfigure total."""
    c = _mk_cluster(num_gpus=1)
    lat = []
    try:
        t0 = time.perf_counter()
        for i in range(8):
            t1 = time.perf_counter()
            c.submit_pod(f"cp-{i}", "cpx-1x36")
            c.wait_pod_scheduled(f"cp-{i}")
            lat.append((time.perf_counter() - t1) * 1000)
        wall = time.perf_counter() - t0
        prepared = c.prepared("node-0")
        assert len(prepared) == 8
        reconfigs = len(c.agents["node-0"].reconfigure_events)
    finally:
        c.stop()
    return {
        "scenario": "cpx8",
        "latency": _lat_stats(lat),
        "pods_per_s": round(8 / wall, 2),
        "reconfigures": reconfigs,
    }


def scenario_mixed100(args) -> dict:
    """100 pods, random cpx/qpx mix, 8 GPUs, bin-packed."""
    c = _mk_cluster(num_gpus=8, policy=args.policy)
    rng = random.Random(42)
    lat = []
    placed = 0
    n = args.pods or 100
    try:
        t0 = time.perf_counter()
        live = []
        for i in range(n):
            prof = rng.choice(["cpx-1x36", "qpx-2x72"])
            name = f"mx-{i}"
            t1 = time.perf_counter()
            c.submit_pod(name, prof)
            outcome = c.wait_pod_outcome(name, timeout=10.0)
            while outcome == "unschedulable":
                # out of capacity: drain oldest pods one at a time until the
                # request fits (mixed modes may need a whole GPU to go idle).
                # Wait for the CAPACITY to free (allocation drained), then
                # give the event-driven re-place a short window — the
                # controller reacts to the CR event in single-digit ms.
                if not live:
                    raise RuntimeError(f"{name} unplaceable on empty node")
                old = live.pop(0)
                c.delete_pod(old)
                c.wait_pod_gone(old)
                c.wait_pod_unallocated(old)
                try:
                    c.wait_pod_scheduled(name, timeout=0.25)
                    outcome = "scheduled"
                except TimeoutError:
                    continue
            lat.append((time.perf_counter() - t1) * 1000)
            live.append(name)
            placed += 1
        wall = time.perf_counter() - t0
        frag = fragmentation(c, "node-0")
        reconfigs = len(c.agents["node-0"].reconfigure_events)
    finally:
        c.stop()
    return {
        "scenario": "mixed100",
        "policy": args.policy,
        "pods": placed,
        "pods_per_s": round(placed / wall, 2),
        "latency": _lat_stats(lat),
        "reconfigures": reconfigs,
        "fragmentation": frag,
    }


def scenario_vllm(args) -> dict:
    """vLLM-shaped deployment pod (reference: samples/vllm_dep.yaml requests
    mig-3g.20gb; MI355X analog: a 2-XCD/72GB partition). This scenario is
    SIMULATED (env contract only, FakeAmdSmi); the measured version — a
    real transformer decode loop inside the allocated partition on metal —
    is tests/test_serving_payload.py::test_serving_workload_in_allocated_partition
    (gpu tier)."""
    c = _mk_cluster(num_gpus=8)
    try:
        t0 = time.perf_counter()
        c.submit_pod("vllm-0", "qpx-2x72", namespace="serving")
        c.wait_pod_scheduled("vllm-0", namespace="serving", timeout=10.0)
        dt = (time.perf_counter() - t0) * 1000
        env = c.pod_env("vllm-0", namespace="serving")
        prep = c.prepared("node-0")
        (p,) = prep.values()
        assert p["xcds"] == 2 and p["memoryGB"] == 72
    finally:
        c.stop()
    return {"scenario": "vllm", "alloc_ms": round(dt, 3),
            "partition": {"xcds": 2, "memoryGB": 72},
            "env_keys": sorted(env)}


def scenario_churn(args) -> dict:
    """N create/delete cycles with a bounded live pool across 8 GPUs:
    stresses per-GPU-mode fragmentation and reconfig churn (SURVEY.md §7.3
    'the churn benchmark will stress exactly this')."""
    n = args.pods or 500
    pool_cap = 24
    c = _mk_cluster(num_gpus=8, policy=args.policy,
                    compute_latency_s=args.reconfig_latency)
    rng = random.Random(7)
    lat = []
    live = []
    failures = 0
    try:
        t0 = time.perf_counter()
        for i in range(n):
            if live and (len(live) >= pool_cap or rng.random() < 0.45):
                victim = live.pop(rng.randrange(len(live)))
                c.delete_pod(victim)
                c.wait_pod_gone(victim)
            prof = rng.choice(["cpx-1x36", "qpx-2x72", "dpx-4x144"])
            name = f"ch-{i}"
            t1 = time.perf_counter()
            c.submit_pod(name, prof)
            outcome = c.wait_pod_outcome(name, timeout=10.0)
            if outcome == "scheduled":
                lat.append((time.perf_counter() - t1) * 1000)
                live.append(name)
            else:
                # controller surfaced no-capacity immediately: drop request
                failures += 1
                c.delete_pod(name)
                c.wait_pod_gone(name)
        wall = time.perf_counter() - t0
        frag = fragmentation(c, "node-0")
        reconfigs = len(c.agents["node-0"].reconfigure_events)
        recfg_ms = [e["set_wall_ms"] for e in c.agents["node-0"].reconfigure_events]
    finally:
        c.stop()
    return {
        "scenario": "churn",
        "policy": args.policy,
        "cycles": n,
        "unplaceable": failures,
        "pods_per_s": round(n / wall, 2),
        "latency": _lat_stats(lat),
        "reconfigures": reconfigs,
        "reconfigure_ms": _lat_stats(recfg_ms),
        "fragmentation": frag,
    }



def scenario_flatlat(args) -> dict:
    """Allocation latency vs live-partition count: the north-star requires
    latency to stay FLAT as partition count grows (cached enumeration; the
    reference re-inits NVML per reconcile and degrades). Fill 8 GPUs with 64
    CPX pods, bucketing latency by load level."""
    c = _mk_cluster(num_gpus=8)
    buckets = {}
    try:
        for i in range(64):
            t0 = time.perf_counter()
            c.submit_pod(f"fl-{i}", "cpx-1x36")
            c.wait_pod_scheduled(f"fl-{i}", timeout=15.0)
            dt = (time.perf_counter() - t0) * 1000
            buckets.setdefault(i // 16, []).append(dt)  # 0-15, 16-31, ...
    finally:
        c.stop()
    out = {"scenario": "flatlat",
           "p50_by_load": {f"pods_{16*k}_{16*k+15}":
                           round(statistics.median(v), 3)
                           for k, v in sorted(buckets.items())}}
    p50s = list(out["p50_by_load"].values())
    out["flatness_max_over_min"] = round(max(p50s) / min(p50s), 2)
    return out


def scenario_manynode(args) -> dict:
    """Cluster-scale placement: 32 nodes x 8 GPUs (256 GPUs, 2048 CPX
    slots), 400 mixed-profile pods with no node pins. Stresses the
    controller's per-admission scan over all node CRs and the cross-node
    argmax; reports placement balance across nodes."""
    n_nodes, n_pods = 32, args.pods or 400
    c = _mk_cluster(num_gpus=8, nodes=n_nodes, policy=args.policy)
    rng = random.Random(7)
    lat = []
    try:
        t0 = time.perf_counter()
        for i in range(n_pods):
            prof = rng.choice(["cpx-1x36", "qpx-2x72"])
            t1 = time.perf_counter()
            c.submit_pod(f"mn-{i}", prof)
            c.wait_pod_scheduled(f"mn-{i}", timeout=30.0)
            lat.append((time.perf_counter() - t1) * 1000)
        wall = time.perf_counter() - t0
        per_node = {}
        for cr in c.store.list("Instaslice"):
            n = len(cr["spec"].get("allocations") or {})
            if n:
                per_node[cr["metadata"]["name"]] = n
    finally:
        c.stop()
    counts = sorted(per_node.values())
    return {
        "scenario": "manynode",
        "policy": args.policy,
        "nodes": n_nodes,
        "pods": n_pods,
        "pods_per_s": round(n_pods / wall, 2),
        "latency": _lat_stats(lat),
        "nodes_used": len(per_node),
        "allocs_per_used_node": {"min": counts[0], "max": counts[-1]}
        if counts else {},
    }


def scenario_preempt(args) -> dict:
    """Preemption latency: node full of low-priority pods; measure
    time-to-scheduled for high-priority arrivals (eviction + drain +
    nominated re-place, org.instaslice/priority)."""
    from instaslice_amd.api.types import new_pod

    c = _mk_cluster(num_gpus=1)
    lat = []
    try:
        for i in range(8):
            c.store.create(new_pod(f"lo-{i}", profile="cpx-1x36", priority=1))
            c.wait_pod_scheduled(f"lo-{i}")
        for i in range(args.pods or 8):
            t0 = time.perf_counter()
            c.store.create(new_pod(f"hi-{i}", profile="cpx-1x36",
                                   priority=10))
            c.wait_pod_scheduled(f"hi-{i}", timeout=30.0)
            lat.append((time.perf_counter() - t0) * 1000)
    finally:
        c.stop()

    # plan 2: whole-GPU preemption — an SPX pod displaces 8 low-priority
    # CPX pods (evict all + nominate GPU + drain + mode flip + place)
    c = _mk_cluster(num_gpus=2)
    lat2 = []
    try:
        from instaslice_amd.api.types import new_pod

        for rnd in range(args.pods or 5):
            for i in range(8):
                c.store.create(new_pod(f"v{rnd}-{i}", profile="cpx-1x36",
                                       priority=1))
                c.wait_pod_scheduled(f"v{rnd}-{i}", timeout=15.0)
            t0 = time.perf_counter()
            c.store.create(new_pod(f"big{rnd}", profile="spx-8x288",
                                   priority=10))
            c.wait_pod_scheduled(f"big{rnd}", timeout=30.0)
            lat2.append((time.perf_counter() - t0) * 1000)
            c.delete_pod(f"big{rnd}")
            c.wait_pod_gone(f"big{rnd}")
            c.wait_pod_unallocated(f"big{rnd}")
    finally:
        c.stop()
    return {"scenario": "preempt",
            "preemption_latency": _lat_stats(lat),
            "whole_gpu_preemption_latency": _lat_stats(lat2)}


SCENARIOS = {
    "plumbing": scenario_plumbing,
    "manynode": scenario_manynode,
    "preempt": scenario_preempt,
    "flatlat": scenario_flatlat,
    "cpx8": scenario_cpx8,
    "mixed100": scenario_mixed100,
    "vllm": scenario_vllm,
    "churn": scenario_churn,
}


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--scenario", default="all", choices=["all", *SCENARIOS])
    ap.add_argument("--pods", type=int, default=0, help="override pod count")
    ap.add_argument("--policy", default="packed-fit")
    ap.add_argument("--reconfig-latency", type=float, default=0.0,
                    help="simulated seconds per compute-mode set (churn)")
    args = ap.parse_args(argv)
    names = list(SCENARIOS) if args.scenario == "all" else [args.scenario]
    results = []
    for name in names:
        res = SCENARIOS[name](args)
        results.append(res)
        print(json.dumps(res), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())

