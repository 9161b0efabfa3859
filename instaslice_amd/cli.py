"""Operational entry points — the analog of the reference's two binaries
(cmd/controller/main.go, cmd/daemonset/main.go) plus operator conveniences.

    python -m instaslice_amd controlplane [--port 7080] [--shards K]
                                        [--data STATE] [--policy packed-fit]
    python -m instaslice_amd store      [--port 7080] [--native] [--data STATE]
    python -m instaslice_amd controller --store HOST:PORT [--policy packed-fit]
                                        [--metrics-port 8080] [--grace 30]
                                        [--leader-elect]
                                        [--shard-index I --shard-count K]
    python -m instaslice_amd daemonset  --store HOST:PORT [--node-name NAME]
                                        [--fake N] [--metrics-port 8084]
                                        [--reset-on-empty]
    python -m instaslice_amd submit     --store HOST:PORT --name P --profile PR
                                        [--node NODE] [--wait]
    python -m instaslice_amd delete     --store HOST:PORT --name P
    python -m instaslice_amd status     --store HOST:PORT
    python -m instaslice_amd describe   --store HOST:PORT --name P
    python -m instaslice_amd top        --store HOST:PORT [--interval S]
    python -m instaslice_amd cordon     --store HOST:PORT --node N [--uncordon]
    python -m instaslice_amd payload    [info|vecadd|membw|busy|census] ...

NODE_NAME env is honored for the daemonset (downward-API parity with
config/manager/manager.yaml's fieldRef env).
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import socket
import sys
import time

from instaslice_amd.utils import get_logger

log = get_logger("cli")


def _connect(store_arg: str, reconnect: bool = False):
    from instaslice_amd.store.netstore import NetStoreClient

    host, _, port = store_arg.rpartition(":")
    return NetStoreClient(host or "127.0.0.1", int(port), reconnect=reconnect)


def _wait_forever():
    stop = []
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    signal.signal(signal.SIGINT, lambda *a: stop.append(1))
    while not stop:
        time.sleep(0.5)


def cmd_store(args) -> int:
    from instaslice_amd.store.memstore import MemStore
    from instaslice_amd.store.native import NativeStoreServer, stored_available
    from instaslice_amd.store.netstore import StoreServer

    backing = None
    use_native = (args.native or (args.native is None and stored_available()))
    if use_native:
        server = NativeStoreServer(port=args.port,
                                   persist_path=args.data).start()
        log.info("native store (instaslice-stored) serving on %s:%d (data=%s)",
                 server.host, server.port, args.data or "in-memory")
    else:
        backing = MemStore(persist_path=args.data) if args.data else None
        server = StoreServer(store=backing, port=args.port).start()
        log.info("store serving on %s:%d (data=%s)", server.host, server.port,
                 args.data or "in-memory")
    _wait_forever()
    server.stop()
    if backing:
        backing.close()
    return 0


def cmd_controlplane(args) -> int:
    """Store + sharded controllers in one command (runtime/controlplane):
    the single-node quick-start and bench topology."""
    import multiprocessing as mp

    from instaslice_amd.runtime.controlplane import run_control_plane

    ctx = mp.get_context("spawn")
    parent_conn, child_conn = ctx.Pipe()
    proc = ctx.Process(
        target=run_control_plane, args=(child_conn, args.policy),
        kwargs={
            "teardown_grace_s": args.grace,
            "port": args.port,
            "persist_path": args.data,
            "controller_shards": args.shards,
        },
        name="control-plane",
    )
    proc.start()
    port = parent_conn.recv()
    log.info("control plane on 127.0.0.1:%d (%d controller shard(s))",
             port, max(1, args.shards))
    try:
        _wait_forever()
    finally:
        try:
            parent_conn.send("stop")
        except (BrokenPipeError, OSError):
            pass
        proc.join(timeout=5.0)
        if proc.is_alive():
            proc.terminate()
    return 0


def cmd_controller(args) -> int:
    from instaslice_amd.controller.reconciler import PodController
    from instaslice_amd.metrics import get_metrics, serve_http

    store = _connect(args.store, reconnect=True)
    lease = None
    if args.leader_elect:
        from instaslice_amd.runtime.lease import LeaderLease

        # one lease PER SHARD: replicas of the same shard compete, distinct
        # shards run concurrently (a single shared lease would serialize the
        # whole sharded deployment down to one controller)
        lease_name = ("instaslice-controller"
                      if args.shard_count <= 1
                      else f"instaslice-controller-shard-{args.shard_index}")
        lease = LeaderLease(store, lease_name).start()
        log.info("waiting for leadership of %s...", lease_name)
        lease.wait_leader()
    controller = PodController(store, policy=args.policy,
                               teardown_grace_s=args.grace,
                               shard_index=args.shard_index,
                               shard_count=args.shard_count)
    controller.start()
    srv = None
    if args.metrics_port:
        srv = serve_http(get_metrics(), args.metrics_port)
        log.info("metrics on :%d", args.metrics_port)
    log.info("controller running (policy=%s grace=%.0fs)", args.policy, args.grace)
    _wait_forever()
    controller.stop()
    if srv:
        srv.shutdown()
    if lease:
        lease.stop()
    return 0


def cmd_daemonset(args) -> int:
    from instaslice_amd.agent.daemonset import NodeAgent
    from instaslice_amd.metrics import get_metrics, serve_http

    node = args.node_name or os.environ.get("NODE_NAME") or socket.gethostname()
    store = _connect(args.store, reconnect=True)
    if args.fake:
        from instaslice_amd.smi.fake import FakeAmdSmi

        smi = FakeAmdSmi(num_gpus=args.fake, node_name=node)
    else:
        from instaslice_amd.smi.native import NativeAmdSmi

        smi = NativeAmdSmi()
    agent = NodeAgent(store, smi, node, reset_mode_on_empty=args.reset_on_empty)
    agent.start()
    plugin = None
    if getattr(args, "devplugin", False):
        from instaslice_amd.devplugin import DevicePluginShim

        plugin = DevicePluginShim(store, smi, node,
                                  resource=args.devplugin_resource).start()
        log.info("device-plugin shim advertising %s", args.devplugin_resource)
    srv = None
    if args.metrics_port:
        srv = serve_http(get_metrics(), args.metrics_port)
        log.info("metrics on :%d", args.metrics_port)
    log.info("daemonset running on node %s (%s)", node,
             "fake" if args.fake else "amdsmi")
    _wait_forever()
    if plugin:
        plugin.stop()
    agent.stop()
    if srv:
        srv.shutdown()
    return 0


def cmd_submit(args) -> int:
    from instaslice_amd.api.types import new_pod

    store = _connect(args.store)
    sel = {"kubernetes.io/hostname": args.node} if args.node else None
    pod = new_pod(args.name, namespace=args.namespace, profile=args.profile,
                  node_selector=sel)
    store.create(pod)
    print(f"pod {args.namespace}/{args.name} submitted (gated)")
    if args.wait:
        deadline = time.monotonic() + args.timeout
        while time.monotonic() < deadline:
            p = store.get("Pod", args.name, args.namespace)
            if not p["spec"].get("schedulingGates"):
                cm = store.get("ConfigMap", args.name, args.namespace)
                print(f"scheduled; env: {json.dumps(cm['data'])}")
                return 0
            time.sleep(0.05)
        print("timed out waiting for scheduling", file=sys.stderr)
        return 1
    return 0


def cmd_delete(args) -> int:
    store = _connect(args.store)
    store.delete("Pod", args.name, args.namespace)
    print(f"pod {args.namespace}/{args.name} deletion requested")
    return 0


def cmd_status(args) -> int:
    store = _connect(args.store)
    out = {"nodes": [], "pods": []}
    for cr in store.list("Instaslice"):
        spec = cr.get("spec", {})
        out["nodes"].append({
            "node": cr["metadata"]["name"],
            "cordoned": bool(spec.get("cordoned")),
            "heartbeatAgeS": (
                round(time.time() - float(cr["status"]["heartbeat"]), 1)
                if (cr.get("status") or {}).get("heartbeat") else None
            ),
            "gpus": {
                u[:8]: {
                    "mode": f"{g.get('computeMode')}/{g.get('memoryMode')}",
                    "used": g.get("usedOrdinals", []),
                    "modeLocked": g.get("modeLocked", False),
                    "metrics": ((cr.get("status") or {})
                                .get("gpuMetrics", {}).get(u)),
                }
                for u, g in (spec.get("gpus") or {}).items()
            },
            "allocations": {
                a["podName"]: f"{a['profile']}@{a['gpuUUID'][:8]}#{a['ordinal']} "
                              f"({a['allocationStatus']})"
                for a in (spec.get("allocations") or {}).values()
            },
            "prepared": len(spec.get("prepared") or {}),
        })
    for pod in store.list("Pod"):
        out["pods"].append({
            "name": pod["metadata"]["name"],
            "gated": bool(pod["spec"].get("schedulingGates")),
            "deleting": bool(pod["metadata"].get("deletionTimestamp")),
        })
    print(json.dumps(out, indent=2))
    return 0


def cmd_cordon(args) -> int:
    """Mark a node unschedulable for NEW placements (running partitions are
    untouched) — kubectl-cordon analog; `uncordon` reverses it."""
    store = _connect(args.store)
    want = not args.uncordon
    store.patch("Instaslice", args.node, "instaslice-system", [
        {"op": "set", "path": ["spec", "cordoned"], "value": want},
    ], quiet=True)
    print(f"node {args.node} {'cordoned' if want else 'uncordoned'}")
    store.close()
    return 0


def cmd_describe(args) -> int:
    """Pod detail + its Events (kubectl-describe analog)."""
    store = _connect(args.store)
    out = {}
    try:
        pod = store.get("Pod", args.name, args.namespace)
        out["pod"] = {
            "name": pod["metadata"]["name"],
            "namespace": pod["metadata"].get("namespace", ""),
            "gated": bool(pod["spec"].get("schedulingGates")),
            "annotations": pod["metadata"].get("annotations") or {},
            "limits": (pod["spec"]["containers"][0]["resources"]
                       .get("limits", {})),
        }
    except Exception as e:  # noqa: BLE001
        out["pod"] = f"not found: {e}"
    for cr in store.list("Instaslice"):
        for a in (cr["spec"].get("allocations") or {}).values():
            if a["podName"] == args.name and a["namespace"] == args.namespace:
                out["allocation"] = a
    out["events"] = sorted(
        (
            {"reason": e["reason"], "type": e["type"], "count": e["count"],
             "message": e["message"], "lastTimestamp": e["lastTimestamp"]}
            for e in store.list("Event", args.namespace)
            if e.get("involvedObject", {}).get("name") == args.name
        ),
        key=lambda e: e["lastTimestamp"],
    )
    print(json.dumps(out, indent=2))
    store.close()
    return 0


def cmd_top(args) -> int:
    """Live cluster view driven by store watches (kubectl-get -w analog):
    one line per state change, plus a periodic occupancy summary."""
    store = _connect(args.store)
    w = store.watch(replay=True, filters=[{"kind": "Pod"},
                                          {"kind": "Instaslice"}])
    last_summary = 0.0
    try:
        while True:
            ev = w.next(timeout=1.0)
            now = time.monotonic()
            if ev is not None:
                et, obj = ev
                md = obj["metadata"]
                if obj["kind"] == "Pod":
                    gates = "gated" if obj["spec"].get("schedulingGates") else "ungated"
                    print(f"{et:9s} pod {md.get('namespace','')}/{md['name']} {gates}",
                          flush=True)
                else:
                    spec = obj.get("spec", {})
                    allocs = spec.get("allocations") or {}
                    print(f"{et:9s} instaslice {md['name']} "
                          f"allocs={len(allocs)} "
                          f"prepared={len(spec.get('prepared') or {})}",
                          flush=True)
            if now - last_summary > args.interval:
                last_summary = now
                for cr in store.list("Instaslice"):
                    spec = cr.get("spec", {})
                    used = sum(len(g.get("usedOrdinals", []))
                               for g in (spec.get("gpus") or {}).values())
                    total = sum(
                        {"SPX": 1, "DPX": 2, "TPX": 3, "QPX": 4, "CPX": 8}
                        .get(g.get("computeMode", "SPX"), 1)
                        for g in (spec.get("gpus") or {}).values())
                    print(f"--- {cr['metadata']['name']}: {used}/{total} "
                          f"partitions used, "
                          f"{len(spec.get('allocations') or {})} allocations",
                          flush=True)
    except KeyboardInterrupt:
        pass
    finally:
        w.stop()
        store.close()
    return 0


def cmd_payload(args) -> int:
    import subprocess

    bin_path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "bin", "instaslice-payload")
    return subprocess.call([bin_path] + args.payload_args)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="instaslice-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("store", help="run the state store server")
    p.add_argument("--port", type=int, default=7080)
    p.add_argument("--data", default=None,
                   help="persist state to this file (checkpoint/resume; "
                        "JSON under the Python server, msgpack under the "
                        "native daemon)")
    p.add_argument("--native", action="store_true", default=None,
                   help="serve with the C++ instaslice-stored daemon "
                        "(default: auto when built and --data is unset)")
    p.add_argument("--no-native", dest="native", action="store_false",
                   help="force the Python store server")
    p.set_defaults(fn=cmd_store)

    p = sub.add_parser("controlplane",
                       help="store + sharded controllers in one command")
    p.add_argument("--port", type=int, default=7080)
    p.add_argument("--policy", default="packed-fit")
    p.add_argument("--grace", type=float, default=30.0)
    p.add_argument("--shards", type=int, default=1,
                   help="controller shard processes")
    p.add_argument("--data", default=None,
                   help="persist state (checkpoint/resume)")
    p.set_defaults(fn=cmd_controlplane)

    p = sub.add_parser("controller", help="run the cluster controller")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--policy", default="packed-fit")
    p.add_argument("--grace", type=float, default=30.0,
                   help="teardown grace seconds (reference: 30)")
    p.add_argument("--metrics-port", type=int, default=8080)
    p.add_argument("--leader-elect", action="store_true")
    p.add_argument("--shard-index", type=int, default=0,
                   help="this controller's shard (crc32 pod ownership)")
    p.add_argument("--shard-count", type=int, default=1,
                   help="total controller shards (run one process per shard)")
    p.set_defaults(fn=cmd_controller)

    p = sub.add_parser("daemonset", help="run the per-node agent")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--node-name", default=None)
    p.add_argument("--fake", type=int, default=0,
                   help="use FakeAmdSmi with N GPUs instead of libamd_smi")
    p.add_argument("--metrics-port", type=int, default=8084)
    p.add_argument("--reset-on-empty", action="store_true",
                   help="return drained GPUs to SPX/NPS1 (reference parity)")
    p.add_argument("--devplugin", action="store_true",
                   help="also run the device-plugin shim (re-advertises "
                        "partition devices on node capacity after flips)")
    p.add_argument("--devplugin-resource", default="amd.com/gpu",
                   help="extended-resource name the shim advertises")
    p.set_defaults(fn=cmd_daemonset)

    p = sub.add_parser("submit", help="submit a gated pod requesting a partition")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--name", required=True)
    p.add_argument("--namespace", default="default")
    p.add_argument("--profile", required=True, help="e.g. cpx-1x36")
    p.add_argument("--node", default=None)
    p.add_argument("--wait", action="store_true")
    p.add_argument("--timeout", type=float, default=60.0)
    p.set_defaults(fn=cmd_submit)

    p = sub.add_parser("delete", help="delete a pod (starts teardown)")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--name", required=True)
    p.add_argument("--namespace", default="default")
    p.set_defaults(fn=cmd_delete)

    p = sub.add_parser("status", help="cluster state summary")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.set_defaults(fn=cmd_status)

    p = sub.add_parser("cordon", help="drain a node for new placements")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--node", required=True)
    p.add_argument("--uncordon", action="store_true")
    p.set_defaults(fn=cmd_cordon)

    p = sub.add_parser("describe", help="pod detail + events")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--name", required=True)
    p.add_argument("--namespace", default="default")
    p.set_defaults(fn=cmd_describe)

    p = sub.add_parser("top", help="live cluster view (watch-driven)")
    p.add_argument("--store", default="127.0.0.1:7080")
    p.add_argument("--interval", type=float, default=10.0)
    p.set_defaults(fn=cmd_top)

    p = sub.add_parser("payload", help="run the HIP payload binary")
    p.add_argument("payload_args", nargs="*", default=["info"])
    p.set_defaults(fn=cmd_payload)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
