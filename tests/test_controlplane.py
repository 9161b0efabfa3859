"""run_control_plane process topologies: single (co-located controller) and
sharded (store-only + K controller shard processes), over whichever store
backend is available. This is the exact topology bench.py rank 0 spawns."""

import multiprocessing as mp
import time

import pytest

from instaslice_amd.agent.daemonset import NodeAgent
from instaslice_amd.api.types import new_pod
from instaslice_amd.runtime.controlplane import run_control_plane
from instaslice_amd.smi.fake import FakeAmdSmi
from instaslice_amd.store.netstore import NetStoreClient


@pytest.mark.parametrize("shards", [1, 2])
def test_control_plane_process_topology(shards):
    ctx = mp.get_context("spawn")
    parent, child = ctx.Pipe()
    proc = ctx.Process(target=run_control_plane, args=(child,),
                       kwargs={"controller_shards": shards})
    proc.start()
    agent = store = None
    try:
        assert parent.poll(60), "control plane never reported its port"
        port = parent.recv()
        store = NetStoreClient("127.0.0.1", port)
        agent = NodeAgent(store, FakeAmdSmi(num_gpus=2, node_name="node-0"),
                          "node-0", heartbeat_every_s=0)
        agent.start()
        for i in range(8):
            store.create(new_pod(f"cp{i}", profile="cpx-1x36"))
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            pods = store.list("Pod")
            if len(pods) == 8 and all(
                not p["spec"].get("schedulingGates") for p in pods
            ):
                break
            time.sleep(0.05)
        pods = store.list("Pod")
        assert all(not p["spec"].get("schedulingGates") for p in pods)
        cr = store.get("Instaslice", "node-0", "instaslice-system")
        slots = {(a["gpuUUID"], a["ordinal"])
                 for a in cr["spec"]["allocations"].values()}
        assert len(slots) == 8
    finally:
        if agent:
            agent.stop()
        if store:
            store.close()
        try:
            parent.send("stop")
        except (BrokenPipeError, OSError):
            pass
        proc.join(timeout=10)
        if proc.is_alive():
            proc.terminate()
