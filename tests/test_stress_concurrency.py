"""Concurrency stress: prepare/unprepare/repartition/health/publish racing
on one DeviceState — the Python analog of the reference's `go test -race`
coverage, exercising the per-claim + per-GPU locking model.
"""

import random
import threading

import pytest

from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.cdi.handler import CDIHandler
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.sharing.shared import SharedComputeManager
from k8s_dra_driver_amd.state.checkpoint import CheckpointStore
from k8s_dra_driver_amd.state.devicestate import DeviceState, PrepareError


def make_claim(uid, devices, configs=None):
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
                        for dev in devices
                    ],
                    "config": configs or [],
                }
            }
        },
    }


@pytest.mark.timeout(120)
def test_stress_mixed_operations(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    state = DeviceState(
        lib,
        CDIHandler(cdi_root=str(tmp_path / "cdi")),
        CheckpointStore(str(tmp_path / "ckpt")),
        pool_name="n",
        shared_manager=SharedComputeManager(
            root=str(tmp_path / "shared"), use_tmpfs=False
        ),
    )
    errors = []
    expected = (PrepareError,)
    rng = random.Random(42)

    part_cfg = {
        "source": "FromClaim",
        "requests": [],
        "opaque": {
            "driver": "gpu.amd.com",
            "parameters": {
                "apiVersion": API_GROUP_VERSION,
                "kind": "PartitionConfig",
                "computePartition": "CPX",
                "memoryPartition": "NPS1",
                "allowDynamicRepartition": True,
            },
        },
    }

    def worker(wid):
        r = random.Random(wid)
        for i in range(30):
            gpu = r.randrange(8)
            uid = f"w{wid}-i{i}"
            op = r.random()
            try:
                if op < 0.7:
                    # normal lifecycle on a whole GPU
                    state.prepare(make_claim(uid, [f"gpu-{gpu}"]))
                    state.unprepare(uid)
                elif op < 0.85:
                    # dynamic repartition attempt (may be refused if held)
                    state.prepare(
                        make_claim(uid, [f"gpu-{gpu}"], configs=[part_cfg])
                    )
                    state.unprepare(uid)
                else:
                    # idempotent double prepare + double unprepare
                    c = make_claim(uid, [f"gpu-{gpu}"])
                    state.prepare(c)
                    state.prepare(c)
                    state.unprepare(uid)
                    state.unprepare(uid)
            except expected:
                pass  # contention refusals are part of the contract
            except Exception as e:  # real bug
                errors.append((wid, i, repr(e)))

    threads = [threading.Thread(target=worker, args=(w,)) for w in range(12)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()

    assert not errors, errors[:5]
    # steady state: nothing prepared, no leaked CDI specs or checkpoints
    assert state.checkpoints.list_all() == {}
    assert state.cdi.list_claim_spec_uids() == []
    for idx in range(8):
        assert state.claims_holding_gpu(idx) == []


@pytest.mark.timeout(60)
def test_stress_prepare_vs_health_republish(tmp_path):
    """prepare/unprepare racing the health monitor's publish callback."""
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    for i in range(20):
        kube.put_resource_claim(make_claim(f"u{i}", [f"gpu-{i % 8}"]))
    stop = threading.Event()

    def health_loop():
        while not stop.is_set():
            driver.health.check_once()

    t = threading.Thread(target=health_loop)
    t.start()
    try:
        for i in range(20):
            res = driver.node_prepare_resources(
                [ClaimRef("d", f"c-u{i}", f"u{i}")]
            )[f"u{i}"]
            assert not res.error
            driver.node_unprepare_resources([ClaimRef("d", f"c-u{i}", f"u{i}")])
    finally:
        stop.set()
        t.join()
    assert driver.state.checkpoints.list_all() == {}


@pytest.mark.timeout(120)
def test_stress_selfheal_vs_claims_vs_tampering(tmp_path):
    """Round-2 machinery under fire: prepare/unprepare storm while an
    external tamperer deletes/mutates ResourceSlices and the publisher's
    synchronous self-heal watch repairs them — no deadlock (RLock +
    emit-outside-store-lock), no lost slices, locks GC'd."""
    from k8s_dra_driver_amd import DRIVER_NAME
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    errors = []
    stop = threading.Event()

    def claim_storm(tid):
        try:
            for i in range(80):
                uid = f"s{tid}-{i}"
                gpu = (tid * 80 + i) % 8
                kube.put_resource_claim(make_claim(uid, [f"gpu-{gpu}"]))
                driver.state.prepare(make_claim(uid, [f"gpu-{gpu}"]))
                driver.state.unprepare(uid)
        except Exception as e:  # pragma: no cover
            errors.append(f"claims[{tid}]: {e!r}")

    def tamperer():
        rng = random.Random(7)
        while not stop.is_set():
            slices = kube.list_resource_slices(DRIVER_NAME)
            if slices:
                s = rng.choice(slices)
                if rng.random() < 0.5:
                    kube.delete_resource_slice(s["metadata"]["name"])
                else:
                    s["spec"]["devices"] = s["spec"]["devices"][:-1]
                    try:
                        kube.update_resource_slice(s)
                    except Exception:
                        pass

    threads = [
        threading.Thread(target=claim_storm, args=(t,)) for t in range(6)
    ]
    tamper = threading.Thread(target=tamperer, daemon=True)
    for t in threads:
        t.start()
    tamper.start()
    for t in threads:
        t.join(timeout=60)
        assert not t.is_alive(), "claim storm deadlocked"
    stop.set()
    tamper.join(timeout=10)
    assert errors == [], errors[:5]
    # final heal pass: publisher restores the full device set
    driver.publish_resources()
    slices = kube.list_resource_slices(DRIVER_NAME)
    names = sorted(d["name"] for s in slices for d in s["spec"]["devices"])
    assert names == [f"gpu-{i}" for i in range(8)]
    # all claims unprepared: refcounted lock map fully GC'd
    assert driver.state._claim_locks == {}
    assert driver.state.claims_holding_gpu(0) == []
    driver.shutdown(unpublish=False)
