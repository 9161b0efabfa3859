"""RCCL/gloo allreduce payload: CPU (gloo ws=2) here, RCCL on the GPU tier."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent


def _launch(nproc, env_extra=None):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", **(env_extra or {}))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
         "--master-port", "29591", "-m", "instaslice_amd.ops.allreduce_check",
         "--numel", "65536", "--iters", "5", "--backend", "gloo"],
        capture_output=True, text=True, timeout=300, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][0]
    return json.loads(line)


def test_allreduce_gloo_ws2():
    res = _launch(2)
    assert res["ok"] and res["max_err"] == 0.0
    assert res["world"] == 2 and res["backend"] == "gloo"


@pytest.mark.gpu
def test_allreduce_rccl_in_partition():
    """Single rank inside the device set a pod would see: proves torch+RCCL
    initialize against the partition (ws=1 on this 1-GPU pool; the 8-GPU
    driver box exercises ws>1 through bench/scale runs)."""
    out = subprocess.run(
        [sys.executable, "-m", "instaslice_amd.ops.allreduce_check",
         "--numel", "65536"],
        capture_output=True, text=True, timeout=300,
        env=dict(os.environ, ROCR_VISIBLE_DEVICES="0"), cwd=ROOT)
    assert out.returncode == 0, out.stdout + out.stderr
    res = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert res["ok"] and res["device"].startswith("cuda")
