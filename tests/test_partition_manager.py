"""Dynamic repartition engine tests: drain guards, mode sequencing."""

import pytest

from k8s_dra_driver_amd.partition.manager import PartitionManager, RepartitionRefused


def test_noop_when_already_in_mode(fake_lib):
    mgr = PartitionManager(fake_lib)
    assert mgr.ensure_mode(0, "SPX", "NPS1", allow_dynamic=False) is False


def test_switch_requires_allow_dynamic(fake_lib):
    mgr = PartitionManager(fake_lib)
    with pytest.raises(RepartitionRefused, match="allowDynamicRepartition"):
        mgr.ensure_mode(0, "CPX", "NPS4", allow_dynamic=False)


def test_switch_to_cpx_nps4(fake_lib):
    mgr = PartitionManager(fake_lib)
    assert mgr.ensure_mode(0, "CPX", "NPS4", allow_dynamic=True) is True
    g0 = fake_lib.enumerate()[0]
    assert (g0.compute_partition, g0.memory_partition) == ("CPX", "NPS4")
    assert len(g0.partitions) == 8


def test_reverse_switch_bridges_through_nps1(fake_lib):
    mgr = PartitionManager(fake_lib)
    mgr.ensure_mode(0, "CPX", "NPS4", allow_dynamic=True)
    # CPX/NPS4 -> SPX/NPS1 requires NPS1 before the compute switch; the
    # engine must sequence it (raw HAL would refuse SPX under NPS4).
    assert mgr.ensure_mode(0, "SPX", "NPS1", allow_dynamic=True) is True
    g0 = fake_lib.enumerate()[0]
    assert (g0.compute_partition, g0.memory_partition) == ("SPX", "NPS1")


def test_drain_guard_blocks_held_gpu(fake_lib):
    holders = {1: ["other-claim"]}
    mgr = PartitionManager(fake_lib, in_use_fn=lambda i: holders.get(i, []))
    with pytest.raises(RepartitionRefused, match="other prepared claim"):
        mgr.ensure_mode(1, "CPX", "NPS1", allow_dynamic=True, requesting_claim="me")
    # the requesting claim itself does not block
    holders[1] = ["me"]
    assert mgr.ensure_mode(1, "CPX", "NPS1", allow_dynamic=True, requesting_claim="me")


def test_invalid_combo_rejected_before_touching_hw(fake_lib):
    mgr = PartitionManager(fake_lib)
    with pytest.raises(ValueError):
        mgr.ensure_mode(0, "SPX", "NPS4", allow_dynamic=True)
    assert fake_lib.faults.call_counts.get("set_compute_partition", 0) == 0
