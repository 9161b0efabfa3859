"""Partition model / profile catalog unit tests."""

import pytest

from instaslice_amd.partition import (
    ComputeMode,
    MemoryMode,
    catalog_from_amdsmi_profiles,
    extract_profile_from_limits,
    mi355x_catalog,
    parse_profile_name,
    xcd_mask,
)


def test_mi355x_catalog_shapes():
    cat = mi355x_catalog()
    names = {p.name for p in cat.profiles}
    assert names == {"spx-8x288", "dpx-4x144", "qpx-2x72", "cpx-1x36"}
    cpx = cat.by_name("cpx-1x36")
    assert cpx.partitions_per_gpu == 8
    assert cpx.memory_gb == 36
    assert cpx.preferred_memory is MemoryMode.NPS4
    spx = cat.by_name("spx-8x288")
    assert spx.partitions_per_gpu == 1
    assert spx.memory_gb == 288


def test_catalog_roundtrip():
    cat = mi355x_catalog()
    cat2 = type(cat).from_dict(cat.to_dict())
    assert [p.name for p in cat2.profiles] == [p.name for p in cat.profiles]


def test_parse_profile_name():
    mode, xcds, gb = parse_profile_name("qpx-2x72")
    assert mode is ComputeMode.QPX and xcds == 2 and gb == 72
    with pytest.raises(ValueError):
        parse_profile_name("1g.5gb")  # NVIDIA MIG names are not ours
    with pytest.raises(ValueError):
        parse_profile_name("cpx-1x")


def test_extract_profile_from_limits():
    assert (
        extract_profile_from_limits(
            {"amd.com/cpx-1x36": 1, "org.instaslice/pod-a": 1, "cpu": 2}
        )
        == "cpx-1x36"
    )
    assert extract_profile_from_limits({"cpu": 1}) is None
    with pytest.raises(ValueError):
        extract_profile_from_limits({"amd.com/cpx-1x36": 1, "amd.com/qpx-2x72": 1})


def test_xcd_mask():
    assert xcd_mask(0, 1) == 0b1
    assert xcd_mask(7, 1) == 0b10000000
    assert xcd_mask(1, 2) == 0b1100
    assert xcd_mask(1, 4) == 0b11110000
    assert xcd_mask(0, 8) == 0xFF


def test_catalog_from_discovery():
    raw = [
        {"profile_type": "SPX", "num_partitions": 1, "profile_index": 0,
         "memory_caps": ["NPS1"]},
        {"profile_type": "CPX", "num_partitions": 8, "profile_index": 3,
         "memory_caps": ["NPS1", "NPS4"]},
        {"profile_type": "WEIRD", "num_partitions": 5, "profile_index": 9,
         "memory_caps": []},  # unknown future mode: skipped, not fatal
    ]
    cat = catalog_from_amdsmi_profiles("AMD Instinct MI355X", 288, raw)
    names = {p.name for p in cat.profiles}
    assert names == {"spx-8x288", "cpx-1x36"}
    assert cat.by_name("cpx-1x36").profile_index == 3
    assert cat.by_name("cpx-1x36").preferred_memory is MemoryMode.NPS4


def test_catalog_discovery_empty_falls_back():
    cat = catalog_from_amdsmi_profiles("AMD Instinct MI355X", 288, [])
    assert {p.name for p in cat.profiles} == {
        "spx-8x288", "dpx-4x144", "qpx-2x72", "cpx-1x36"
    }
