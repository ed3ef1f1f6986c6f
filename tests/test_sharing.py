"""Sharing tests: time-slicing manager + shared-compute supervisor."""

import pytest

from k8s_dra_driver_amd.api.types import SharedComputeSettings, TimeSlicingSettings
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.model import AllocatableDevice
from k8s_dra_driver_amd.sharing.shared import (
    SharedComputeManager,
    cu_mask_hex,
)
from k8s_dra_driver_amd.sharing.timeslice import SharingError, TimeSlicingManager


def _gpu_devices(lib, *indices):
    gpus = lib.enumerate()
    return [AllocatableDevice.from_gpu(gpus[i]) for i in indices]


class TestTimeSlicing:
    def test_sets_quantum_per_parent_gpu(self, fake_lib):
        mgr = TimeSlicingManager(fake_lib)
        touched = mgr.set_timeslice(
            _gpu_devices(fake_lib, 0, 1), TimeSlicingSettings(interval="Short")
        )
        assert touched == [0, 1]
        assert fake_lib.get_timeslice_quantum(0) == 1000
        assert fake_lib.get_timeslice_quantum(1) == 1000

    def test_restore_default(self, fake_lib):
        mgr = TimeSlicingManager(fake_lib)
        mgr.set_timeslice(_gpu_devices(fake_lib, 2), TimeSlicingSettings(interval="Long"))
        mgr.restore_default([2])
        assert fake_lib.get_timeslice_quantum(2) is None

    def test_partitions_rejected(self, fake_lib):
        fake_lib.set_compute_partition(0, "CPX")
        g0 = fake_lib.enumerate()[0]
        part = AllocatableDevice.from_partition(g0, g0.partitions[0])
        with pytest.raises(SharingError, match="whole GPUs only"):
            TimeSlicingManager(fake_lib).set_timeslice(
                [part], TimeSlicingSettings(interval="Short")
            )


class TestCuMask:
    def test_mask_values(self):
        assert cu_mask_hex(0, 4, 8) == "0x0f"
        assert cu_mask_hex(4, 4, 8) == "0xf0"
        m = cu_mask_hex(0, 64, 256)
        assert int(m, 16) == (1 << 64) - 1


class TestSharedCompute:
    def _mgr(self, tmp_path):
        return SharedComputeManager(root=str(tmp_path / "shared"), use_tmpfs=False)

    def test_session_env_and_mounts(self, tmp_path, fake_lib):
        mgr = self._mgr(tmp_path)
        devs = _gpu_devices(fake_lib, 0)
        s = mgr.start_session(
            "claim-uid-1",
            devs,
            SharedComputeSettings(
                default_memory_limit="8Gi", default_cu_share_percent=25
            ),
        )
        assert any(e.startswith("HSA_CU_MASK=0:") for e in s.env)
        assert any("AMD_DRA_MEMORY_LIMIT" in e for e in s.env)
        assert s.mounts[0].container_path == "/dev/shm"
        edits = s.container_edits()
        assert edits.mounts and edits.env

    def test_disjoint_cu_ranges_across_claims(self, tmp_path, fake_lib):
        mgr = self._mgr(tmp_path)
        devs = _gpu_devices(fake_lib, 0)
        settings = SharedComputeSettings(default_cu_share_percent=25)
        s1 = mgr.start_session("claim-a", devs, settings)
        s2 = mgr.start_session("claim-b", devs, settings)
        m1 = int(next(e for e in s1.env if "HSA_CU_MASK" in e).split(":")[1], 16)
        m2 = int(next(e for e in s2.env if "HSA_CU_MASK" in e).split(":")[1], 16)
        assert m1 & m2 == 0  # disjoint CU slices
        assert bin(m1).count("1") == 64  # 25% of 256

    def test_capacity_exhaustion(self, tmp_path, fake_lib):
        mgr = self._mgr(tmp_path)
        devs = _gpu_devices(fake_lib, 0)
        settings = SharedComputeSettings(default_cu_share_percent=50)
        mgr.start_session("a", devs, settings)
        mgr.start_session("b", devs, settings)
        with pytest.raises(RuntimeError, match="no CU capacity"):
            mgr.start_session("c", devs, settings)
        mgr.stop_session("a")
        mgr.start_session("c", devs, settings)  # freed capacity is reusable

    def test_stop_idempotent_and_cleans_dir(self, tmp_path, fake_lib):
        mgr = self._mgr(tmp_path)
        s = mgr.start_session(
            "claim-x", _gpu_devices(fake_lib, 1), SharedComputeSettings()
        )
        import os

        assert os.path.isdir(s.shm_dir)
        mgr.stop_session(s.session_id)
        assert not os.path.isdir(s.shm_dir)
        mgr.stop_session(s.session_id)  # no-op

    def test_recover_session_restores_cu_bookkeeping(self, tmp_path, fake_lib):
        mgr = self._mgr(tmp_path)
        devs = _gpu_devices(fake_lib, 0)
        settings = SharedComputeSettings(default_cu_share_percent=50)
        s1 = mgr.start_session("a", devs, settings)
        # new manager (plugin restart) recovers from checkpointed session
        mgr2 = self._mgr(tmp_path)
        mgr2.recover_session(s1)
        s2 = mgr2.start_session("b", devs, settings)
        m1 = int(next(e for e in s1.env if "HSA_CU_MASK" in e).split(":")[1], 16)
        m2 = int(next(e for e in s2.env if "HSA_CU_MASK" in e).split(":")[1], 16)
        assert m1 & m2 == 0


class TestTimeSlicingHonesty:
    """advisory time-slicing is surfaced, not silent."""

    def _ts_cfg(self, interval="Short"):
        return {
            "source": "FromClaim",
            "requests": [],
            "opaque": {
                "driver": "gpu.amd.com",
                "parameters": {
                    "apiVersion": "resource.gpu.amd.com/v1alpha1",
                    "kind": "GpuConfig",
                    "sharing": {
                        "strategy": "TimeSlicing",
                        "timeSlicingConfig": {"interval": interval},
                    },
                },
            },
        }

    def _state(self, tmp_path, lib):
        from k8s_dra_driver_amd.cdi.handler import CDIHandler
        from k8s_dra_driver_amd.state.checkpoint import CheckpointStore
        from k8s_dra_driver_amd.state.devicestate import DeviceState

        return DeviceState(
            lib,
            CDIHandler(cdi_root=str(tmp_path / "cdi")),
            CheckpointStore(str(tmp_path / "ckpt")),
            pool_name="n",
        )

    def _claim(self, uid, dev_cfgs):
        results = [
            {
                "request": f"req-{i}",
                "driver": "gpu.amd.com",
                "pool": "n",
                "device": dev,
            }
            for i, (dev, _) in enumerate(dev_cfgs)
        ]
        configs = []
        for i, (_, interval) in enumerate(dev_cfgs):
            c = self._ts_cfg(interval)
            c["requests"] = [f"req-{i}"]
            configs.append(c)
        return {
            "metadata": {"namespace": "d", "name": f"c-{uid}", "uid": uid},
            "status": {
                "allocation": {
                    "devices": {"results": results, "config": configs}
                }
            },
        }

    def test_advisory_timeslice_emits_warning(self, tmp_path):
        from k8s_dra_driver_amd.hal import FakeDeviceLib

        lib = FakeDeviceLib()
        lib.open()
        lib.timeslice_effective_flag = False  # a real amdgpu pool
        state = self._state(tmp_path, lib)
        warnings = []
        state.on_warning = lambda info, reason, msg: warnings.append(
            (info.uid, reason, msg)
        )
        state.prepare(self._claim("u-adv", [("gpu-0", "Short")]))
        assert len(warnings) == 1
        uid, reason, msg = warnings[0]
        assert uid == "u-adv" and reason == "TimeSlicingAdvisory"
        assert "ADVISORY" in msg and "timeSlicingEffective" in msg

    def test_effective_timeslice_no_warning(self, tmp_path):
        from k8s_dra_driver_amd.hal import FakeDeviceLib

        lib = FakeDeviceLib()
        lib.open()
        state = self._state(tmp_path, lib)
        warnings = []
        state.on_warning = lambda *a: warnings.append(a)
        state.prepare(self._claim("u-eff", [("gpu-0", "Short")]))
        assert warnings == []

    def test_timeslice_effective_attribute_published(self):
        from k8s_dra_driver_amd.hal import FakeDeviceLib
        from k8s_dra_driver_amd.hal.model import AllocatableDevice

        lib = FakeDeviceLib()
        lib.open()
        lib.timeslice_effective_flag = False
        dev = AllocatableDevice.from_gpu(lib.enumerate()[0]).to_device()
        attrs = dev["basic"]["attributes"]
        assert attrs["gpu.amd.com/timeSlicingEffective"]["bool"] is False
        assert attrs["gpu.amd.com/repartitionCapable"]["bool"] is True

    def test_mixed_intervals_different_gpus_apply_per_group(self, tmp_path):
        from k8s_dra_driver_amd.hal import FakeDeviceLib

        lib = FakeDeviceLib()
        lib.open()
        state = self._state(tmp_path, lib)
        state.prepare(
            self._claim("u-mix", [("gpu-0", "Short"), ("gpu-1", "Long")])
        )
        assert lib.get_timeslice_quantum(0) == 1000  # Short
        assert lib.get_timeslice_quantum(1) == 10000  # Long
        state.unprepare("u-mix")
        assert lib.get_timeslice_quantum(0) is None
        assert lib.get_timeslice_quantum(1) is None

    def test_conflicting_intervals_same_gpu_rejected(self, tmp_path):
        import pytest

        from k8s_dra_driver_amd.hal import FakeDeviceLib
        from k8s_dra_driver_amd.state.devicestate import PrepareError

        lib = FakeDeviceLib()
        lib.open()
        state = self._state(tmp_path, lib)
        with pytest.raises(PrepareError, match="conflicting TimeSlicing"):
            state.prepare(
                self._claim("u-con", [("gpu-0", "Short"), ("gpu-0", "Long")])
            )
