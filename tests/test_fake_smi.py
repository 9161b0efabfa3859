"""FakeAmdSmi behavior tests: the device model the integration tier rests on."""

import pytest

from instaslice_amd.smi import FakeAmdSmi, SmiBusy, SmiError, SmiNotSupported


@pytest.fixture
def smi():
    s = FakeAmdSmi(num_gpus=2)
    s.init()
    return s


def test_enumeration_spx_default(smi):
    gpus = smi.list_gpus()
    assert len(gpus) == 2
    for g in gpus:
        assert g.compute_mode == "SPX" and g.memory_mode == "NPS1"
        assert g.memory_gb == 288
        assert len(g.partitions) == 1
        assert g.partitions[0].memory_gb == 288


def test_cpx_reenumeration_and_device_index_packing(smi):
    gpus = smi.list_gpus()
    smi.set_compute_partition(gpus[0].uuid, "CPX")
    fresh = smi.list_gpus()
    assert len(fresh[0].partitions) == 8
    assert [p.ordinal for p in fresh[0].partitions] == list(range(8))
    assert fresh[0].partitions[0].memory_gb == 36
    # node-wide HIP indices re-pack: gpu1's single SPX device moves to index 8
    assert [p.device_index for p in fresh[0].partitions] == list(range(8))
    assert fresh[1].partitions[0].device_index == 8
    # partition uuids change across mode generations
    assert fresh[0].partitions[0].uuid != gpus[0].partitions[0].uuid


def test_uuid_determinism_across_instances():
    a, b = FakeAmdSmi(num_gpus=1), FakeAmdSmi(num_gpus=1)
    a.init(); b.init()
    ga, gb = a.list_gpus()[0], b.list_gpus()[0]
    assert ga.uuid == gb.uuid
    a.set_compute_partition(ga.uuid, "QPX")
    b.set_compute_partition(gb.uuid, "QPX")
    assert [p.uuid for p in a.list_gpus()[0].partitions] == [
        p.uuid for p in b.list_gpus()[0].partitions
    ]


def test_busy_gpu_refuses_mode_change(smi):
    g = smi.list_gpus()[0]
    smi.mark_busy(g.uuid, g.partitions[0].uuid)
    with pytest.raises(SmiBusy):
        smi.set_compute_partition(g.uuid, "CPX")
    smi.mark_busy(g.uuid, g.partitions[0].uuid, busy=False)
    smi.set_compute_partition(g.uuid, "CPX")
    assert smi.get_compute_partition(g.uuid) == "CPX"


def test_memory_mode_validity(smi):
    g = smi.list_gpus()[0]
    with pytest.raises(SmiNotSupported):
        smi.set_memory_partition(g.uuid, "NPS4")  # illegal under SPX
    smi.set_compute_partition(g.uuid, "CPX")
    smi.set_memory_partition(g.uuid, "NPS4")
    assert smi.get_memory_partition(g.uuid) == "NPS4"
    # flipping back to SPX forces memory mode legal again
    smi.set_compute_partition(g.uuid, "SPX")
    assert smi.get_memory_partition(g.uuid) == "NPS1"


def test_tpx_not_supported_on_mi355x(smi):
    g = smi.list_gpus()[0]
    with pytest.raises(SmiNotSupported):
        smi.set_compute_partition(g.uuid, "TPX")


def test_requires_init():
    s = FakeAmdSmi()
    with pytest.raises(SmiError):
        s.list_gpus()


def test_profile_config(smi):
    g = smi.list_gpus()[0]
    cfg = smi.get_profile_config(g.uuid)
    types = {c["profile_type"]: c for c in cfg}
    assert set(types) == {"SPX", "DPX", "QPX", "CPX"}
    assert types["CPX"]["num_partitions"] == 8
    assert "NPS4" in types["CPX"]["memory_caps"]


def test_fault_injection_hook(smi):
    calls = []

    def hook(verb, gpu):
        calls.append(verb)
        if verb == "set_compute_partition":
            raise SmiError("injected")

    smi.fault_hook = hook
    g = smi.list_gpus()[0]
    with pytest.raises(SmiError, match="injected"):
        smi.set_compute_partition(g.uuid, "CPX")
    assert "set_compute_partition" in calls
