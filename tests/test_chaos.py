"""Chaos soak: random HAL fault injection under concurrent claim churn.

The §5.3 (failure detection / fault injection) stress artifact: with a
randomized fraction of HAL calls failing mid-operation, every claim must
either complete its full lifecycle or fail CLEANLY — and when the storm
ends, the node must be exactly as it started: no GPU holders, no shared
sessions, no CU-range bookkeeping, no claim CDI specs, no checkpoints,
every GPU back in SPX. Rollback correctness, not just crash survival.
"""

import random
import threading

import pytest

from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.cdi.handler import CDIHandler
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.base import HalError
from k8s_dra_driver_amd.sharing.shared import SharedComputeManager
from k8s_dra_driver_amd.state.checkpoint import CheckpointStore
from k8s_dra_driver_amd.state.devicestate import DeviceState, PrepareError

#: HAL ops the storm may fail (prepare/unprepare path touchpoints)
FAULTABLE_OPS = [
    "set_timeslice_quantum",
    "set_compute_partition",
    "set_memory_partition",
    "device_node_paths",
]


class _ChaosInjector:
    """Randomly fails a fraction of fire() calls (thread-safe enough:
    races only change WHICH call fails, which is the point)."""

    def __init__(self, rate: float, seed: int):
        self.rate = rate
        self.rng = random.Random(seed)
        self.lock = threading.Lock()
        self.call_counts = {}
        self.fired = 0

    def fail_next(self, *a, **kw):  # FaultInjector interface
        pass

    def set_latency(self, *a, **kw):
        pass

    def fire(self, op: str) -> None:
        with self.lock:
            self.call_counts[op] = self.call_counts.get(op, 0) + 1
            if op in FAULTABLE_OPS and self.rng.random() < self.rate:
                self.fired += 1
                raise HalError(f"chaos: injected failure in {op}")


def _claim(uid, dev, flavor):
    cfgs = []
    if flavor == "ts":
        cfgs = [
            {
                "source": "FromClaim",
                "requests": [],
                "opaque": {
                    "driver": "gpu.amd.com",
                    "parameters": {
                        "apiVersion": API_GROUP_VERSION,
                        "kind": "GpuConfig",
                        "sharing": {
                            "strategy": "TimeSlicing",
                            "timeSlicingConfig": {"interval": "Short"},
                        },
                    },
                },
            }
        ]
    elif flavor == "shared":
        cfgs = [
            {
                "source": "FromClaim",
                "requests": [],
                "opaque": {
                    "driver": "gpu.amd.com",
                    "parameters": {
                        "apiVersion": API_GROUP_VERSION,
                        "kind": "GpuConfig",
                        "sharing": {
                            "strategy": "SharedCompute",
                            "sharedComputeConfig": {
                                "defaultCuSharePercent": 25
                            },
                        },
                    },
                },
            }
        ]
    elif flavor == "carve":
        cfgs = [
            {
                "source": "FromClaim",
                "requests": [],
                "opaque": {
                    "driver": "gpu.amd.com",
                    "parameters": {
                        "apiVersion": API_GROUP_VERSION,
                        "kind": "PartitionConfig",
                        "computePartition": "CPX",
                        "memoryPartition": "NPS4",
                        "allowDynamicRepartition": True,
                    },
                },
            }
        ]
    return {
        "metadata": {"namespace": "d", "name": f"c-{uid}", "uid": uid},
        "status": {
            "allocation": {
                "devices": {
                    "results": [
                        {
                            "request": "gpu",
                            "driver": "gpu.amd.com",
                            "pool": "n",
                            "device": dev,
                        }
                    ],
                    "config": cfgs,
                }
            }
        },
    }


@pytest.mark.timeout(180)
@pytest.mark.parametrize("seed", [1, 7])
def test_chaos_storm_leaves_node_clean(tmp_path, seed):
    chaos = _ChaosInjector(rate=0.08, seed=seed)
    lib = FakeDeviceLib(faults=chaos)
    lib.open()
    shared = SharedComputeManager(
        root=str(tmp_path / "shared"), use_tmpfs=False
    )
    state = DeviceState(
        lib,
        CDIHandler(cdi_root=str(tmp_path / "cdi")),
        CheckpointStore(str(tmp_path / "ckpt")),
        pool_name="n",
        shared_manager=shared,
    )
    stats = {"ok": 0, "failed": 0}
    stats_lock = threading.Lock()
    errors = []

    def worker(tid):
        rng = random.Random(1000 + tid)
        try:
            for i in range(60):
                uid = f"c{tid}-{i}"
                gpu = rng.randrange(8)
                flavor = rng.choice(
                    ["plain", "ts", "shared", "carve", "autocarve"]
                )
                if flavor == "autocarve":
                    # scheduler-driven: claim a (possibly prospective)
                    # partition directly; prepare carves on demand
                    dev = f"gpu-{gpu}-cpx-{rng.randrange(8)}"
                    claim = _claim(uid, dev, "plain")
                else:
                    claim = _claim(uid, f"gpu-{gpu}", flavor)
                try:
                    state.prepare(claim)
                except PrepareError:
                    with stats_lock:
                        stats["failed"] += 1
                    # a failed prepare must leave nothing behind for
                    # this uid (rollback contract)
                    assert state.checkpoints.read(uid) is None
                    assert uid not in state.cdi.list_claim_spec_uids()
                    continue
                with stats_lock:
                    stats["ok"] += 1
                # unprepare until it sticks (kubelet retry semantics:
                # a failed unprepare is retried and must be idempotent)
                for _ in range(10):
                    try:
                        state.unprepare(uid)
                        break
                    except Exception:
                        continue
                else:
                    raise AssertionError(f"unprepare of {uid} never stuck")
        except Exception as e:  # pragma: no cover
            errors.append(f"worker {tid}: {e!r}")

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
        assert not t.is_alive(), "chaos storm deadlocked"
    assert errors == [], errors[:3]
    assert chaos.fired > 0, "the storm never injected anything"
    assert stats["ok"] > 0 and stats["failed"] > 0, stats

    # final sweep: deferred restores may need a drained retry
    state.unprepare("final-sweep-noop")

    # --- the node must be exactly as it started -----------------------
    for i in range(8):
        assert state.claims_holding_gpu(i) == [], f"gpu-{i} still held"
    assert state._claim_locks == {}
    assert shared._sessions == {}
    for st in shared._gpu_state.values():
        assert st.ranges == {}, "leaked CU ranges"
    assert state.cdi.list_claim_spec_uids() == []
    assert state.checkpoints.list_all() == {}
    for g in lib.enumerate():
        assert (g.compute_partition, g.memory_partition) == ("SPX", "NPS1"), (
            f"gpu-{g.index} left as {g.compute_partition}/{g.memory_partition}"
        )
    print(
        f"chaos[{seed}]: {stats['ok']} ok, {stats['failed']} failed-clean, "
        f"{chaos.fired} faults injected"
    )
