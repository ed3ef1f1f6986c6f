"""DeviceState end-to-end tests on the fake HAL: the prepare/unprepare
state machine with checkpoints, CDI, sharing and dynamic repartition.
"""

import os
import threading

import pytest

from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.cdi.handler import CDIHandler
from k8s_dra_driver_amd.cdi.spec import read_spec_file
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.sharing.shared import SharedComputeManager
from k8s_dra_driver_amd.state.checkpoint import CheckpointStore
from k8s_dra_driver_amd.state.devicestate import DeviceState, PrepareError

POOL = "test-node"


def make_state(tmp_path, lib=None):
    lib = lib or FakeDeviceLib()
    lib.open()
    cdi = CDIHandler(cdi_root=str(tmp_path / "cdi"))
    cps = CheckpointStore(str(tmp_path / "ckpt"))
    shared = SharedComputeManager(root=str(tmp_path / "shared"), use_tmpfs=False)
    state = DeviceState(
        lib, cdi, cps, pool_name=POOL, shared_manager=shared
    )
    return state, lib


def make_claim(uid, devices, configs=None, requests=None):
    """Build a v1beta1-shaped ResourceClaim with an allocation."""
    results = []
    for i, dev in enumerate(devices):
        results.append(
            {
                "request": (requests or [f"req-{i}" for i in range(len(devices))])[i],
                "driver": "gpu.amd.com",
                "pool": POOL,
                "device": dev,
            }
        )
    return {
        "metadata": {"namespace": "default", "name": f"claim-{uid}", "uid": uid},
        "status": {
            "allocation": {
                "devices": {"results": results, "config": configs or []}
            }
        },
    }


def opaque(params, source="FromClaim", requests=None):
    return {
        "source": source,
        "requests": requests or [],
        "opaque": {"driver": "gpu.amd.com", "parameters": params},
    }


class TestPrepareBasic:
    def test_prepare_whole_gpu(self, tmp_path):
        state, lib = make_state(tmp_path)
        devs = state.prepare(make_claim("uid-1", ["gpu-0"]))
        assert len(devs) == 1
        d = devs[0]
        assert d["device_name"] == "gpu-0"
        assert d["pool_name"] == POOL
        assert d["cdi_device_ids"] == [
            "k8s.gpu.amd.com/device=gpu-0",
            "k8s.gpu.amd.com/claim=uid-1-gpu-0",
        ]
        # claim CDI spec exists with claim env
        assert "uid-1" in state.cdi.list_claim_spec_uids()

    def test_prepare_is_idempotent(self, tmp_path):
        state, lib = make_state(tmp_path)
        claim = make_claim("uid-2", ["gpu-1"])
        first = state.prepare(claim)
        second = state.prepare(claim)
        assert first == second

    def test_prepare_unknown_device_fails(self, tmp_path):
        state, _ = make_state(tmp_path)
        with pytest.raises(PrepareError, match="not found"):
            state.prepare(make_claim("uid-3", ["gpu-99"]))

    def test_prepare_no_allocation_fails(self, tmp_path):
        state, _ = make_state(tmp_path)
        with pytest.raises(PrepareError, match="allocation"):
            state.prepare({"metadata": {"uid": "u"}, "status": {}})

    def test_multi_gpu_claim(self, tmp_path):
        state, _ = make_state(tmp_path)
        devs = state.prepare(
            make_claim("uid-4", ["gpu-0", "gpu-1", "gpu-2", "gpu-3"])
        )
        assert [d["device_name"] for d in devs] == [
            "gpu-0",
            "gpu-1",
            "gpu-2",
            "gpu-3",
        ]

    def test_unprepare_removes_everything(self, tmp_path):
        state, _ = make_state(tmp_path)
        state.prepare(make_claim("uid-5", ["gpu-0"]))
        state.unprepare("uid-5")
        assert state.checkpoints.read("uid-5") is None
        assert state.cdi.list_claim_spec_uids() == []
        state.unprepare("uid-5")  # idempotent no-op

    def test_gpu_holders_tracked(self, tmp_path):
        state, _ = make_state(tmp_path)
        state.prepare(make_claim("uid-6", ["gpu-2"]))
        assert state.claims_holding_gpu(2) == ["uid-6"]
        state.unprepare("uid-6")
        assert state.claims_holding_gpu(2) == []


class TestSharingConfigs:
    def test_time_slicing_applied_and_restored(self, tmp_path):
        state, lib = make_state(tmp_path)
        cfg = opaque(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "GpuConfig",
                "sharing": {
                    "strategy": "TimeSlicing",
                    "timeSlicingConfig": {"interval": "Short"},
                },
            }
        )
        state.prepare(make_claim("uid-ts", ["gpu-0"], configs=[cfg]))
        assert lib.get_timeslice_quantum(0) == 1000
        state.unprepare("uid-ts")
        assert lib.get_timeslice_quantum(0) is None

    def test_shared_compute_session(self, tmp_path):
        state, lib = make_state(tmp_path)
        cfg = opaque(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "GpuConfig",
                "sharing": {
                    "strategy": "SharedCompute",
                    "sharedComputeConfig": {"defaultCuSharePercent": 25},
                },
            }
        )
        state.prepare(make_claim("uid-sc", ["gpu-0"], configs=[cfg]))
        # claim CDI spec carries the session env + shm mount
        spec = read_spec_file(
            os.path.join(
                state.cdi.cdi_root, "k8s.gpu.amd.com-claim-uid-sc.json"
            )
        )
        edits = spec["devices"][0]["containerEdits"]
        assert any("HSA_CU_MASK" in e for e in edits["env"])
        assert edits["mounts"][0]["containerPath"] == "/dev/shm"
        state.unprepare("uid-sc")
        assert state.shared_manager.get_session("uid-sc"[:36]) is None

    def test_invalid_config_rejected(self, tmp_path):
        state, _ = make_state(tmp_path)
        cfg = opaque(
            {"apiVersion": API_GROUP_VERSION, "kind": "GpuConfig", "junk": 1}
        )
        with pytest.raises(PrepareError, match="invalid opaque config"):
            state.prepare(make_claim("uid-bad", ["gpu-0"], configs=[cfg]))

    def test_class_vs_claim_precedence(self, tmp_path):
        state, lib = make_state(tmp_path)
        class_cfg = opaque(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "GpuConfig",
                "sharing": {
                    "strategy": "TimeSlicing",
                    "timeSlicingConfig": {"interval": "Long"},
                },
            },
            source="FromClass",
        )
        claim_cfg = opaque(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "GpuConfig",
                "sharing": {
                    "strategy": "TimeSlicing",
                    "timeSlicingConfig": {"interval": "Short"},
                },
            },
            source="FromClaim",
        )
        state.prepare(
            make_claim("uid-prec", ["gpu-0"], configs=[class_cfg, claim_cfg])
        )
        assert lib.get_timeslice_quantum(0) == 1000  # claim (Short) wins


class TestDynamicRepartition:
    def _partition_cfg(self, compute="CPX", memory="NPS4", allow=True):
        return opaque(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "PartitionConfig",
                "computePartition": compute,
                "memoryPartition": memory,
                "allowDynamicRepartition": allow,
            }
        )

    def test_whole_gpu_claim_with_partition_config(self, tmp_path):
        state, lib = make_state(tmp_path)
        devs = state.prepare(
            make_claim("uid-p1", ["gpu-0"], configs=[self._partition_cfg()])
        )
        # the GPU was carved; the claim holds all 8 CPX partitions
        assert len(devs) == 8
        assert devs[0]["device_name"] == "gpu-0-cpx-0"
        g0 = lib.enumerate()[0]
        assert g0.compute_partition == "CPX"
        assert g0.memory_partition == "NPS4"
        # base CDI spec was rewritten with the partition devices
        base = read_spec_file(
            os.path.join(state.cdi.cdi_root, "k8s.gpu.amd.com-device.json")
        )
        assert any(d["name"] == "gpu-0-cpx-7" for d in base["devices"])

    def test_unprepare_restores_previous_mode(self, tmp_path):
        state, lib = make_state(tmp_path)
        state.prepare(
            make_claim("uid-p2", ["gpu-0"], configs=[self._partition_cfg()])
        )
        state.unprepare("uid-p2")
        g0 = lib.enumerate()[0]
        assert (g0.compute_partition, g0.memory_partition) == ("SPX", "NPS1")

    def test_repartition_refused_without_allow(self, tmp_path):
        state, _ = make_state(tmp_path)
        with pytest.raises(PrepareError, match="allowDynamicRepartition"):
            state.prepare(
                make_claim(
                    "uid-p3",
                    ["gpu-0"],
                    configs=[self._partition_cfg(allow=False)],
                )
            )

    def test_repartition_refused_when_gpu_held(self, tmp_path):
        state, _ = make_state(tmp_path)
        state.prepare(make_claim("uid-hold", ["gpu-0"]))
        with pytest.raises(PrepareError, match="other prepared claim"):
            state.prepare(
                make_claim("uid-p4", ["gpu-0"], configs=[self._partition_cfg()])
            )

    def test_partition_devices_allocatable_after_carve(self, tmp_path):
        """BASELINE config #4 flow: carve, then bind 1:1 claims."""
        state, lib = make_state(tmp_path)
        # carve via a dedicated claim, then release it (mode persists if
        # restore is refused — here restore succeeds, so carve via manager)
        state.partition_manager.ensure_mode(0, "CPX", "NPS4", allow_dynamic=True)
        state.refresh_allocatable()
        names = [d.canonical_name for d in state.allocatable_devices()]
        assert "gpu-0-cpx-5" in names and "gpu-1" in names
        devs = state.prepare(make_claim("uid-p5", ["gpu-0-cpx-5"]))
        assert devs[0]["device_name"] == "gpu-0-cpx-5"
        assert state.claims_holding_gpu(0) == ["uid-p5"]


class TestRecovery:
    def test_restart_recovers_prepared_claims(self, tmp_path):
        state, lib = make_state(tmp_path)
        claim = make_claim("uid-r1", ["gpu-0"])
        first = state.prepare(claim)
        # simulate plugin restart: new DeviceState over the same dirs
        state2 = DeviceState(
            lib,
            state.cdi,
            CheckpointStore(str(tmp_path / "ckpt")),
            pool_name=POOL,
            shared_manager=state.shared_manager,
        )
        assert state2.claims_holding_gpu(0) == ["uid-r1"]
        # idempotent prepare returns the checkpointed devices
        assert state2.prepare(claim) == first

    def test_concurrent_prepares_do_not_serialize_on_one_lock(self, tmp_path):
        state, _ = make_state(tmp_path)
        errs = []

        def run(i):
            try:
                state.prepare(make_claim(f"uid-c{i}", [f"gpu-{i}"]))
            except Exception as e:  # pragma: no cover
                errs.append(e)

        threads = [threading.Thread(target=run, args=(i,)) for i in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert not errs
        assert len(state.checkpoints.list_all()) == 8


class TestOtherPartitionModes:
    def test_dpx_and_qpx_prepare(self, tmp_path):
        """DPX (2x4 XCD) and QPX (4x2 XCD) carve + 1:1 partition claims."""
        state, lib = make_state(tmp_path)
        state.partition_manager.ensure_mode(0, "DPX", "NPS1", allow_dynamic=True)
        state.partition_manager.ensure_mode(1, "QPX", "NPS4", allow_dynamic=True)
        state.refresh_allocatable()
        names = {d.canonical_name for d in state.allocatable_devices()}
        assert {"gpu-0-dpx-0", "gpu-0-dpx-1"} <= names
        assert {"gpu-1-qpx-0", "gpu-1-qpx-3"} <= names
        devs = state.prepare(make_claim("uid-dpx", ["gpu-0-dpx-1"]))
        assert devs[0]["device_name"] == "gpu-0-dpx-1"
        devs = state.prepare(make_claim("uid-qpx", ["gpu-1-qpx-2"]))
        assert devs[0]["device_name"] == "gpu-1-qpx-2"
        # capacities reflect the mode
        qpx = next(
            d for d in state.allocatable_devices()
            if d.canonical_name == "gpu-1-qpx-2"
        ).to_device()
        assert qpx["basic"]["capacity"]["gpu.amd.com/computeUnits"]["value"] == "64"
        assert qpx["basic"]["capacity"]["gpu.amd.com/memory"]["value"] == "72Gi"
