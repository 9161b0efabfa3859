"""CLI + leader-election tests: the full operational topology as processes
would run it (store server, controller, fake daemonset, submit/status)."""

import json
import subprocess
import sys
import time

import pytest

from instaslice_amd.runtime.lease import LeaderLease
from instaslice_amd.store.memstore import MemStore
from instaslice_amd.store.netstore import NetStoreClient, StoreServer


def test_leader_election_single_winner():
    store = MemStore()
    a = LeaderLease(store, "ctl", identity="a", ttl_s=1.0, renew_every_s=0.1).start()
    b = LeaderLease(store, "ctl", identity="b", ttl_s=1.0, renew_every_s=0.1).start()
    try:
        assert a.wait_leader(2.0) or b.wait_leader(2.0)
        time.sleep(0.3)
        assert a.is_leader.is_set() != b.is_leader.is_set()  # exactly one
        winner, loser = (a, b) if a.is_leader.is_set() else (b, a)
        winner.stop()  # releases
        assert loser.wait_leader(3.0), "successor never acquired"
    finally:
        a.stop()
        b.stop()


def test_cli_end_to_end_processes():
    """store + controller + fake daemonset as real subprocesses, driven by
    submit/status/delete — the deployment shape of the k8s manifests."""
    server = StoreServer().start()
    addr = f"127.0.0.1:{server.port}"
    procs = []

    def spawn(*cmd):
        p = subprocess.Popen([sys.executable, "-m", "instaslice_amd", *cmd],
                             stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                             text=True)
        procs.append(p)
        return p

    try:
        spawn("controller", "--store", addr, "--grace", "0", "--metrics-port", "0")
        spawn("daemonset", "--store", addr, "--node-name", "cli-node",
              "--fake", "2", "--metrics-port", "0")
        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "submit", "--store", addr,
             "--name", "clipod", "--profile", "qpx-2x72", "--wait",
             "--timeout", "30"],
            capture_output=True, text=True, timeout=60)
        if out.returncode != 0:
            import json as _json
            diag = [out.stdout, out.stderr,
                    "CRs: " + _json.dumps(server.store.list("Instaslice"))[:1500],
                    "Pods: " + _json.dumps(server.store.list("Pod"))[:800]]
            for p_ in procs:
                p_.terminate()
                try:
                    o, _ = p_.communicate(timeout=5)
                except subprocess.TimeoutExpired:
                    p_.kill()
                    o, _ = p_.communicate()
                diag.append(o[-3000:] if o else "<no output>")
            pytest.fail("submit failed:\n" + "\n=====\n".join(diag))
        assert "ROCR_VISIBLE_DEVICES" in out.stdout

        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "status", "--store", addr],
            capture_output=True, text=True, timeout=30)
        st = json.loads(out.stdout)
        assert st["nodes"][0]["node"] == "cli-node"
        assert "clipod" in st["nodes"][0]["allocations"]

        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "delete", "--store", addr,
             "--name", "clipod"], capture_output=True, text=True, timeout=30)
        assert out.returncode == 0
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            crs = server.store.list("Instaslice")
            if crs and not crs[0]["spec"].get("allocations"):
                break
            time.sleep(0.05)
        else:
            pytest.fail("teardown did not drain")
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        server.stop()


def test_cli_native_store_sharded_controllers():
    """Native store daemon + 2 sharded controller processes + fake daemonset,
    all via the CLI — the scale-out deployment shape."""
    from instaslice_amd.store.native import NativeStoreServer, stored_available

    if not stored_available():
        pytest.skip("instaslice-stored not built")
    server = NativeStoreServer().start()
    addr = f"127.0.0.1:{server.port}"
    procs = []

    def spawn(*cmd):
        p = subprocess.Popen([sys.executable, "-m", "instaslice_amd", *cmd],
                             stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                             text=True)
        procs.append(p)
        return p

    try:
        for i in range(2):
            spawn("controller", "--store", addr, "--grace", "0",
                  "--metrics-port", "0", "--shard-index", str(i),
                  "--shard-count", "2")
        spawn("daemonset", "--store", addr, "--node-name", "cli-node",
              "--fake", "2", "--metrics-port", "0")
        # several pods so both shards own at least one with high likelihood
        for k in range(6):
            out = subprocess.run(
                [sys.executable, "-m", "instaslice_amd", "submit", "--store",
                 addr, "--name", f"shpod{k}", "--profile", "cpx-1x36",
                 "--wait", "--timeout", "30"],
                capture_output=True, text=True, timeout=60)
            assert out.returncode == 0, out.stdout + out.stderr
        client = NetStoreClient("127.0.0.1", server.port)
        cr = client.get("Instaslice", "cli-node", "instaslice-system")
        assert len(cr["spec"]["allocations"]) == 6
        client.close()
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()
        server.stop()


def test_cli_controlplane_one_command():
    """`controlplane` = store + controllers in one process tree."""
    import re

    p = subprocess.Popen(
        [sys.executable, "-m", "instaslice_amd", "controlplane", "--port", "0",
         "--grace", "0", "--shards", "1"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    agent = None
    try:
        port = None
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            line = p.stdout.readline()
            m = re.search(r"control plane on 127\.0\.0\.1:(\d+)", line or "")
            if m:
                port = int(m.group(1))
                break
        assert port, "control plane never reported its port"
        agent = subprocess.Popen(
            [sys.executable, "-m", "instaslice_amd", "daemonset", "--store",
             f"127.0.0.1:{port}", "--node-name", "cp-node", "--fake", "1",
             "--metrics-port", "0"],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "submit", "--store",
             f"127.0.0.1:{port}", "--name", "cppod", "--profile", "spx-8x288",
             "--wait", "--timeout", "30"],
            capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stdout + out.stderr
    finally:
        if agent:
            agent.terminate()
            agent.wait(timeout=5)
        p.terminate()
        p.wait(timeout=10)


def test_cli_top_describe_cordon():
    """Operator commands: top streams events, describe shows pod+events,
    cordon/uncordon gates placement — all over the CLI."""
    from instaslice_amd.store.native import NativeStoreServer, stored_available
    from instaslice_amd.store.netstore import NetStoreClient

    if not stored_available():
        pytest.skip("instaslice-stored not built")
    server = NativeStoreServer().start()
    addr = f"127.0.0.1:{server.port}"
    procs = []

    def spawn(*cmd):
        p = subprocess.Popen([sys.executable, "-m", "instaslice_amd", *cmd],
                             stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                             text=True)
        procs.append(p)
        return p

    try:
        spawn("controller", "--store", addr, "--grace", "0",
              "--metrics-port", "0")
        spawn("daemonset", "--store", addr, "--node-name", "opnode",
              "--fake", "1", "--metrics-port", "0")
        top = spawn("top", "--store", addr, "--interval", "1")
        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "submit", "--store", addr,
             "--name", "oppod", "--profile", "cpx-1x36", "--wait",
             "--timeout", "30"], capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stdout + out.stderr

        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "describe", "--store",
             addr, "--name", "oppod"], capture_output=True, text=True,
            timeout=30)
        desc = json.loads(out.stdout)
        assert desc["pod"]["gated"] is False
        assert desc["allocation"]["profile"] == "cpx-1x36"
        deadline = time.monotonic() + 5
        reasons = set()
        while time.monotonic() < deadline:
            out = subprocess.run(
                [sys.executable, "-m", "instaslice_amd", "describe", "--store",
                 addr, "--name", "oppod"], capture_output=True, text=True,
                timeout=30)
            reasons = {e["reason"] for e in json.loads(out.stdout)["events"]}
            if {"Placed", "PartitionReady"} <= reasons:
                break
            time.sleep(0.1)
        assert {"Placed", "PartitionReady"} <= reasons, reasons

        out = subprocess.run(
            [sys.executable, "-m", "instaslice_amd", "cordon", "--store",
             addr, "--node", "opnode"], capture_output=True, text=True,
            timeout=30)
        assert "cordoned" in out.stdout
        c = NetStoreClient("127.0.0.1", server.port)
        assert c.get("Instaslice", "opnode",
                     "instaslice-system")["spec"]["cordoned"] is True
        c.close()

        top.terminate()
        top_out, _ = top.communicate(timeout=10)
        assert "pod default/oppod" in top_out
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()
        server.stop()


def test_leader_crash_failover_after_ttl():
    """UNGRACEFUL leader death (no release): the successor must acquire
    after the lease TTL expires — the crash-failover path, distinct from
    the graceful-handoff test above."""
    import time

    from instaslice_amd.store.memstore import MemStore

    store = MemStore()
    a = LeaderLease(store, "crash-test", identity="A", ttl_s=0.5,
                    renew_every_s=0.1)
    # A acquires once and then "crashes": no renew loop ever runs
    assert a._try_acquire()
    b = LeaderLease(store, "crash-test", identity="B", ttl_s=0.5,
                    renew_every_s=0.1).start()
    try:
        assert not b.wait_leader(0.3), "B stole a live lease before TTL"
        assert b.wait_leader(3.0), "B never took over after the TTL expired"
        lease = store.get("Lease", "crash-test", "instaslice-system")
        assert lease["spec"]["holderIdentity"] == "B"
    finally:
        b.stop()
