"""Multi-process distributed-path tests (gloo, world_size=2, CPU).

The driver runs bench.py via torch.distributed.run with one rank per GPU at
round end; this tier proves that exact launch path works with no GPU: two
ranks, two fake nodes, one controller on rank 0, TCP store, max-over-ranks
timing, pooled latency samples.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def _run_bench(nproc: int, extra=()):
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", "29581",
        str(ROOT / "bench.py"),
        "--gpus", str(nproc), "--steps", "10", "--warmup", "2", "--fake",
        *extra,
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         env=env, cwd=ROOT)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected one JSON line, got: {lines}"
    return json.loads(lines[0])


def test_bench_two_ranks_gloo():
    res = _run_bench(2)
    assert res["metric"] == "pods_scheduled_per_s"
    assert res["n_gpus"] == 2
    assert res["steps"] == 10
    assert res["value"] > 0
    assert res["scaling"] == "weak"
    import re
    assert re.fullmatch(r"\d+-controller-shards\+2-node-agents",
                        res["config"]["parallelism"])
    assert res["config"]["backend"] == "fake"
    assert res["p50_alloc_latency_ms"] is not None
    # pooled latency samples from both ranks: 2 * 10 recorded steps
    assert res["config"]["global_batch"] == 20


def test_bench_single_rank_no_torchrun():
    """The driver's N=1 invocation: plain `python bench.py`."""
    out = subprocess.run(
        [sys.executable, str(ROOT / "bench.py"), "--steps", "10",
         "--warmup", "2", "--fake"],
        capture_output=True, text=True, timeout=300, cwd=ROOT)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    res = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert res["n_gpus"] == 1 and res["value"] > 0
