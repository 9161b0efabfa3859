"""Device-plugin contract test: the agent's
re-advertisement nudge closed-loop with a device-plugin shim.

Reference analog: the NVIDIA plugin label-toggle reload
(instaslice_daemonset.go:474-497) — after a MIG carve the plugin must
re-advertise or the pod's extended-resource request never schedules.
Here: after a compute-mode flip the partition device population changes;
the agent bumps org.instaslice/last-reconfigure; the shim re-enumerates
and re-advertises amd.com/gpu; only then does the scheduler predicate
pass for the new device count.
"""

import time

import pytest

from instaslice_amd.devplugin import (
    RECONFIGURE_LABEL,
    DevicePluginShim,
    schedulable,
)
from instaslice_amd.runtime.cluster import Cluster
from instaslice_amd.smi import FakeAmdSmi


@pytest.fixture
def rig():
    smi = FakeAmdSmi(num_gpus=2, node_name="node-0")
    c = Cluster(teardown_grace_s=0.0, policy="packed-fit")
    c.add_node("node-0", smi)
    c.start()
    plugin = DevicePluginShim(c.store, smi, "node-0").start()
    yield c, smi, plugin
    plugin.stop()
    c.stop()


def _wait(cond, timeout=10.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if cond():
            return True
        time.sleep(0.01)
    return False


def test_initial_advertisement(rig):
    c, smi, plugin = rig
    # 2 GPUs in SPX = 2 partition devices
    assert _wait(lambda: schedulable(c.store, "node-0", "amd.com/gpu", 2))
    assert not schedulable(c.store, "node-0", "amd.com/gpu", 3)


def test_reconfigure_triggers_readvertisement(rig):
    """A CPX flip turns one GPU into 8 devices: 9 total become schedulable
    ONLY after the agent's label bump reaches the plugin and it
    re-enumerates — the closed loop the reference drives by restarting
    the NVIDIA plugin."""
    c, smi, plugin = rig
    assert _wait(lambda: schedulable(c.store, "node-0", "amd.com/gpu", 2))
    # a pod demanding 9 devices CANNOT schedule yet
    assert not schedulable(c.store, "node-0", "amd.com/gpu", 9)
    before = plugin.advertisements

    c.submit_pod("p1", "cpx-1x36")  # forces one GPU SPX -> CPX
    c.wait_pod_scheduled("p1")
    # the flip happened; the plugin must pick it up via the label bump
    assert _wait(lambda: schedulable(c.store, "node-0", "amd.com/gpu", 9)), (
        "plugin never re-advertised after the mode flip")
    assert plugin.advertisements > before
    node = c.store.get("Node", "node-0", "")
    assert node["metadata"]["labels"].get(RECONFIGURE_LABEL), (
        "agent never bumped the reconfigure label")
    assert node["status"]["capacity"]["amd.com/gpu"] == 9


def test_drain_reverts_advertisement():
    """With reset-on-empty teardown, draining the CPX pod flips the GPU
    back and the plugin re-advertises the shrunken device count."""
    smi = FakeAmdSmi(num_gpus=1, node_name="node-0")
    c = Cluster(teardown_grace_s=0.0, policy="packed-fit",
                reset_mode_on_empty=True)
    c.add_node("node-0", smi)
    c.start()
    plugin = DevicePluginShim(c.store, smi, "node-0").start()
    try:
        assert _wait(lambda: schedulable(c.store, "node-0", "amd.com/gpu", 1))
        c.submit_pod("p1", "cpx-1x36")
        c.wait_pod_scheduled("p1")
        assert _wait(lambda: schedulable(c.store, "node-0", "amd.com/gpu", 8))
        c.delete_pod("p1")
        c.wait_pod_gone("p1")
        assert _wait(lambda: not schedulable(
            c.store, "node-0", "amd.com/gpu", 8)), (
            "advertisement never shrank after the GPU reset to SPX")
        assert schedulable(c.store, "node-0", "amd.com/gpu", 1)
    finally:
        plugin.stop()
        c.stop()
