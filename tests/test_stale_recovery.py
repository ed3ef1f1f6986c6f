"""Stale-hardware recovery: checkpoints re-validated against live devices
at restart (the resume gap SURVEY.md §5.4 flags in the reference)."""

from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
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


def _shared_claim(uid, dev, percent):
    c = claim(uid, dev)
    c["status"]["allocation"]["devices"]["config"] = [
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
                            "defaultCuSharePercent": percent
                        },
                    },
                },
            },
        }
    ]
    return c


def _cu_masks(state, uid):
    """HSA_CU_MASK entries of a claim's checkpoint, as {gpu: int_mask}."""
    pc = state.checkpoints.read(uid)
    out = {}
    for e in pc.claim_env:
        if e.startswith("HSA_CU_MASK="):
            for entry in e.split("=", 1)[1].split(";"):
                idx_s, mask_s = entry.split(":", 1)
                out[int(idx_s)] = int(mask_s, 16)
    return out


def test_restart_preserves_shared_cu_ranges(tmp_path):
    """After a plugin restart, recovered SharedCompute sessions must keep
    their CU-range bookkeeping (parsed back from the checkpointed
    HSA_CU_MASK env) so a new session on the same GPU gets a disjoint
    slice — the spatial-isolation property the recovery path exists to
    preserve (devicestate recovery env=[])."""
    lib = FakeDeviceLib()
    lib.open()
    state = make_state(tmp_path, lib)
    state.prepare(_shared_claim("u-a", "gpu-0", 50))
    masks_a = _cu_masks(state, "u-a")
    assert masks_a and all(m for m in masks_a.values())

    state2 = make_state(tmp_path, lib)  # restart over the same dirs
    state2.prepare(_shared_claim("u-b", "gpu-0", 50))
    masks_b = _cu_masks(state2, "u-b")
    assert masks_b
    for gpu, mb in masks_b.items():
        assert masks_a.get(gpu, 0) & mb == 0, (
            f"gpu-{gpu}: post-restart session overlaps recovered session: "
            f"{masks_a.get(gpu):#x} & {mb:#x}"
        )
