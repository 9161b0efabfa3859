"""Round-trip floor microbench against instaslice-stored: ping, get, patch.
The pod-lifecycle critical path is ~7 such RTs; p50 lifecycle latency can't
beat 7 x RTT. Evidence for DESIGN.md's 'at the architecture floor' claim."""
import statistics
import sys
import time

from instaslice_amd.api.types import new_instaslice
from instaslice_amd.store.native import NativeStoreServer
from instaslice_amd.store.netstore import NetStoreClient

server = NativeStoreServer().start()
c = NetStoreClient("127.0.0.1", server.port)
c.create(new_instaslice("n0"))

def bench(label, fn, n=3000):
    fn()  # warm
    ts = []
    for _ in range(n):
        t0 = time.perf_counter()
        fn()
        ts.append((time.perf_counter() - t0) * 1e6)
    xs = sorted(ts)
    print(f"{label}: p50={xs[len(xs)//2]:.0f}us "
          f"p99={xs[int(0.99*len(xs))]:.0f}us min={xs[0]:.0f}us")

bench("ping          ", lambda: c._call("ping"))
bench("get CR        ", lambda: c.get("Instaslice", "n0", "instaslice-system"))
bench("patch CR quiet", lambda: c.patch(
    "Instaslice", "n0", "instaslice-system",
    [{"op": "set", "path": ["status", "heartbeat"], "value": 1.0}], quiet=True))
c.close()
server.stop()
