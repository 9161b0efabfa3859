"""Serving payload (BASELINE config 4, the vLLM-pod analog): a real
transformer decode loop runs in the partition the operator allocated,
as a child process under the pod's exact env contract — measured, not
simulated."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent


def _run_serving(env_extra=None, **kw):
    cmd = [sys.executable, "-m", "instaslice_amd.ops.serving_check"]
    for k, v in kw.items():
        cmd += [f"--{k}", str(v)]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         env=dict(os.environ, **(env_extra or {})), cwd=ROOT)
    assert out.returncode == 0, out.stdout + out.stderr
    return json.loads(
        [l for l in out.stdout.splitlines() if l.startswith("{")][0])


def test_serving_payload_cpu():
    res = _run_serving(layers=2, dmodel=128, prefill=32, decode=8)
    assert res["ok"] and res["finite"]
    assert res["decode_tokens"] == 8 and res["decode_tok_s"] > 0


@pytest.mark.gpu
def test_serving_workload_in_allocated_partition():
    """Full loop on metal: operator allocates a partition for a
    vllm-shaped pod; the serving payload runs as a child confined by the
    pod ConfigMap's ROCR_VISIBLE_DEVICES; decode must run on cuda in
    bf16 and produce finite logits."""
    from instaslice_amd.partition.profiles import ProfileCatalog
    from instaslice_amd.runtime.cluster import Cluster
    from instaslice_amd.smi.native import NativeAmdSmi

    smi = NativeAmdSmi()
    smi.init()
    gpus = smi.list_gpus()
    assert gpus, "no AMD GPUs"
    g0 = gpus[0]
    c = Cluster(teardown_grace_s=0.0)
    c.add_node("serve-node", smi)
    c.start()
    try:
        cr = c.store.get("Instaslice", "serve-node", "instaslice-system")
        cat = ProfileCatalog.from_dict(cr["spec"]["placements"])
        # the profile matching the live mode (no flip needed: this pool
        # refuses flips — profiles/partition_write_matrix_r02.md)
        profile = next(p for p in cat.profiles
                       if p.compute.value == g0.compute_mode)
        c.submit_pod("vllm-0", profile.name, namespace="serving")
        c.wait_pod_scheduled("vllm-0", namespace="serving", timeout=30.0)
        env = c.pod_env("vllm-0", namespace="serving")
        res = _run_serving(
            env_extra={
                "ROCR_VISIBLE_DEVICES": env["ROCR_VISIBLE_DEVICES"],
                "HIP_VISIBLE_DEVICES": env["HIP_VISIBLE_DEVICES"],
            },
            layers=4, dmodel=512, prefill=128, decode=32)
        print(f"\nserving in partition: {res}")
        assert res["ok"] and res["device"].startswith("cuda")
        assert res["dtype"] == "bfloat16"
        assert res["decode_tok_s"] > 0
        c.delete_pod("vllm-0", namespace="serving")
        c.wait_pod_gone("vllm-0", namespace="serving", timeout=30.0)
    finally:
        c.stop()


def test_payload_serve_protocol_cpu():
    """The serve-mode line protocol itself (ping/quit/unknown) works
    without touching HIP — CPU-tier coverage of the worker loop."""
    bin_path = ROOT / "instaslice_amd" / "bin" / "instaslice-payload"
    if not bin_path.exists():
        pytest.skip("instaslice-payload not built")
    w = subprocess.Popen([str(bin_path), "serve"], stdin=subprocess.PIPE,
                         stdout=subprocess.PIPE, text=True, bufsize=1)
    try:
        def ask(cmd):
            w.stdin.write(cmd + "\n")
            w.stdin.flush()
            return json.loads(w.stdout.readline())

        assert ask("ping")["ok"]
        assert not ask("nonsense")["ok"]
        assert ask("quit")["ok"]
        assert w.wait(timeout=10) == 0
    finally:
        if w.poll() is None:
            w.kill()
