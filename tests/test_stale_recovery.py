"""Stale-hardware recovery: checkpoints re-validated against live devices
at restart (the resume gap SURVEY.md §5.4 flags in the reference)."""

from k8s_dra_driver_amd.cdi.handler import CDIHandler
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.sharing.shared import SharedComputeManager
from k8s_dra_driver_amd.state.checkpoint import CheckpointStore
from k8s_dra_driver_amd.state.devicestate import DeviceState


def make_state(tmp_path, lib):
    return DeviceState(
        lib,
        CDIHandler(cdi_root=str(tmp_path / "cdi")),
        CheckpointStore(str(tmp_path / "ckpt")),
        pool_name="n",
        shared_manager=SharedComputeManager(
            root=str(tmp_path / "shared"), use_tmpfs=False
        ),
    )


def claim(uid, dev):
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
                    ]
                }
            }
        },
    }


def test_restart_after_partition_drift_flags_stale(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    state = make_state(tmp_path, lib)
    state.prepare(claim("u-whole", "gpu-0"))  # prepared against SPX gpu-0
    # plugin "dies"; while down, the GPU is carved to CPX out-of-band
    lib.set_compute_partition(0, "CPX")
    state2 = make_state(tmp_path, lib)  # restart over the same dirs
    assert state2.stale_claims == {"u-whole": ["gpu-0"]}
    # the claim itself is preserved (its pod may still be running)
    assert state2.checkpoints.read("u-whole") is not None
    # and can still be unprepared cleanly
    state2.unprepare("u-whole")
    assert state2.checkpoints.read("u-whole") is None


def test_restart_without_drift_is_clean(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    state = make_state(tmp_path, lib)
    state.prepare(claim("u1", "gpu-1"))
    state2 = make_state(tmp_path, lib)
    assert state2.stale_claims == {}
