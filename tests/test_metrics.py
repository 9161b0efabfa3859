"""Observability tests: metrics recorded by the live stack + HTTP endpoint."""

import urllib.request

from instaslice_amd.metrics import Tracer, get_metrics, serve_http
from instaslice_amd.runtime.cluster import Cluster
from instaslice_amd.smi import FakeAmdSmi


def test_stack_records_metrics():
    m = get_metrics()
    base_created = m.count("allocations_total", ("created",))
    base_deleted = m.count("allocations_total", ("deleted",))
    c = Cluster(teardown_grace_s=0.0)
    c.add_node("mnode", FakeAmdSmi(num_gpus=1, node_name="mnode"))
    c.start()
    try:
        c.submit_pod("mp1", "cpx-1x36")
        c.wait_pod_scheduled("mp1")
        c.delete_pod("mp1")
        c.wait_pod_gone("mp1")
        c.wait_allocations_empty("mnode")
    finally:
        c.stop()
    assert m.count("allocations_total", ("created",)) == base_created + 1
    assert m.count("allocations_total", ("deleted",)) == base_deleted + 1
    assert m.count("partition_reconfigures_total", ("mnode", "CPX/NPS1")) >= 1
    assert m.samples("allocation_latency_seconds")
    assert m.percentile("allocation_latency_seconds", 0.5) is not None
    assert m.samples("reconcile_seconds", ("controller",))


def test_http_endpoint():
    m = get_metrics()
    srv = serve_http(m, 0)
    port = srv.server_address[1]
    try:
        body = urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics").read()
        assert b"instaslice" in body or b"total" in body
        assert urllib.request.urlopen(f"http://127.0.0.1:{port}/healthz").status == 200
        assert urllib.request.urlopen(f"http://127.0.0.1:{port}/readyz").status == 200
    finally:
        srv.shutdown()


def test_tracer_span():
    t = Tracer(capacity=8)
    with t.span("reconcile", engine="test"):
        pass
    for i in range(20):
        t.event("tick", i=i)
    evs = t.dump()
    assert len(evs) == 8  # bounded ring
    assert any(e["kind"] == "tick" and e["i"] == 19 for e in evs)
