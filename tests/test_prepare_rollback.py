"""Prepare-failure rollback: a failed claim must leave the node as it
found it — sharing undone, repartition reverted, no on-disk artifacts."""

import pytest

from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.hal.base import HalError
from tests.test_devicestate import make_claim, make_state, opaque


def _partition_cfg():
    return opaque(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "PartitionConfig",
            "computePartition": "CPX",
            "memoryPartition": "NPS4",
            "allowDynamicRepartition": True,
        }
    )


def _shared_cfg():
    return opaque(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": "SharedCompute",
                "sharedComputeConfig": {"defaultCuSharePercent": 25},
            },
        }
    )


def test_failed_prepare_reverts_repartition(tmp_path):
    state, lib = make_state(tmp_path)
    # carve succeeds, then the base-spec republish after enumeration is
    # fine, but the claim CDI write fails (disk error injected by making
    # the write pool call explode via a poisoned cdi handler)
    orig = state.cdi.create_claim_spec

    def boom(*a, **k):
        raise OSError("disk full")

    state.cdi.create_claim_spec = boom
    from k8s_dra_driver_amd.state.devicestate import PrepareError

    with pytest.raises((PrepareError, OSError)):
        state.prepare(
            make_claim("uid-fail", ["gpu-0"], configs=[_partition_cfg()])
        )
    state.cdi.create_claim_spec = orig
    # the repartition was rolled back and nothing leaked
    g0 = lib.enumerate()[0]
    assert (g0.compute_partition, g0.memory_partition) == ("SPX", "NPS1")
    assert state.checkpoints.read("uid-fail") is None
    assert state.cdi.list_claim_spec_uids() == []
    assert state.claims_holding_gpu(0) == []
    # the claim can be prepared cleanly afterwards
    devs = state.prepare(
        make_claim("uid-fail", ["gpu-0"], configs=[_partition_cfg()])
    )
    assert len(devs) == 8


def test_failed_timeslice_stops_shared_session(tmp_path):
    state, lib = make_state(tmp_path)
    lib.faults.fail_next("set_timeslice_quantum", HalError("broken"))
    ts_cfg = opaque(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": "TimeSlicing",
                "timeSlicingConfig": {"interval": "Short"},
            },
        },
        requests=["req-0"],
    )
    sc_cfg = _shared_cfg()
    sc_cfg["requests"] = ["req-1"]
    from k8s_dra_driver_amd.state.devicestate import PrepareError

    with pytest.raises(PrepareError, match="time-slicing failed"):
        state.prepare(
            make_claim(
                "uid-mix", ["gpu-0", "gpu-1"], configs=[sc_cfg, ts_cfg]
            )
        )
    # the shared session opened for req-1 was rolled back
    assert state.shared_manager.get_session("uid-mix"[:36]) is None
    assert state.checkpoints.read("uid-mix") is None
