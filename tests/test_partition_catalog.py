"""Partition catalog tests: the MIG-profile-matrix analog for gfx950."""

import pytest

from k8s_dra_driver_amd.partition.catalog import (
    COMPUTE_MODES,
    GFX950_VRAM_MIB,
    gfx950_catalog,
    make_profile,
    validate_mode_combo,
)


def test_compute_mode_partition_counts():
    assert COMPUTE_MODES == {"SPX": 1, "DPX": 2, "QPX": 4, "CPX": 8}


def test_cpx_nps4_profile_shape():
    p = make_profile("CPX", "NPS4")
    assert p.num_partitions == 8
    assert p.xcds_per_partition == 1
    assert p.cus_per_partition == 32
    assert p.memory_mib_per_partition == GFX950_VRAM_MIB // 8  # 36 GiB
    assert p.num_memory_domains == 4


def test_memory_domain_binding_cpx_nps4():
    p = make_profile("CPX", "NPS4")
    # 8 partitions pack 2-per-domain over 4 domains
    assert [p.memory_domain_of(i) for i in range(8)] == [0, 0, 1, 1, 2, 2, 3, 3]


def test_memory_slices_cover_stacks_disjointly():
    for cm, mm in [("CPX", "NPS4"), ("QPX", "NPS4"), ("DPX", "NPS1"), ("SPX", "NPS1")]:
        p = make_profile(cm, mm)
        seen = []
        for i in range(p.num_partitions):
            seen.extend(p.memory_slices_of(i))
        assert sorted(seen) == list(range(8)), (cm, mm)


def test_invalid_combos_rejected():
    with pytest.raises(ValueError):
        validate_mode_combo("SPX", "NPS4")
    with pytest.raises(ValueError):
        validate_mode_combo("CPX", "NPS2")
    with pytest.raises(ValueError):
        validate_mode_combo("XPX", "NPS1")
    with pytest.raises(ValueError):
        validate_mode_combo("SPX", "NPS9")


def test_catalog_contains_all_valid_profiles():
    cat = {(p.compute_mode, p.memory_mode) for p in gfx950_catalog()}
    assert ("SPX", "NPS1") in cat
    assert ("CPX", "NPS4") in cat
    assert ("QPX", "NPS4") in cat
    assert ("SPX", "NPS4") not in cat


def test_live_caps_override_static_matrix():
    # hardware reporting NPS1-only must shrink the catalog
    cat = gfx950_catalog(nps_caps_by_mode={"CPX": ("NPS1",)})
    cpx = [p for p in cat if p.compute_mode == "CPX"]
    assert [p.memory_mode for p in cpx] == ["NPS1"]
