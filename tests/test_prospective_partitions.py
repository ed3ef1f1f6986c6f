"""Scheduler-driven dynamic partitioning (DRA partitionable devices,
K8s 1.33 sharedCounters/consumesCounters) — the dynamic-MIG capability
the reference shipped disabled (nvlib.go:560-669), here driven by the
default scheduler: prospective partition devices are published before
any carve, allocation of one triggers an auto-carve in prepare, and the
GPU returns to SPX when its last pod drains."""

import pytest

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import Driver


@pytest.fixture
def driver(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    kube.api_versions = ["v1beta2", "v1beta1"]  # counters need v1beta2+
    d = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        prospective_partitions="cpx",
    )
    d.startup()
    yield d, kube, lib
    d.shutdown(unpublish=False)


def _devices(kube):
    return {
        dev["name"]: dev
        for s in kube.list_resource_slices(DRIVER_NAME)
        for dev in s["spec"]["devices"]
    }


def _claim(uid, dev):
    return {
        "metadata": {"namespace": "d", "name": f"c-{uid}", "uid": uid},
        "status": {
            "allocation": {
                "devices": {
                    "results": [
                        {
                            "request": "gpu",
                            "driver": DRIVER_NAME,
                            "pool": "n",
                            "device": dev,
                        }
                    ]
                }
            }
        },
    }


def test_prospective_devices_published_with_counters(driver):
    d, kube, lib = driver
    devs = _devices(kube)
    # 8 whole GPUs + 8x8 prospective CPX partitions
    assert len(devs) == 8 + 64
    g0 = devs["gpu-0"]
    assert g0["consumesCounters"][0]["counterSet"] == "gpu-0-counters"
    assert len(g0["consumesCounters"][0]["counters"]) == 8
    p3 = devs["gpu-0-cpx-3"]
    assert p3["attributes"]["gpu.amd.com/prospective"]["bool"] is True
    assert p3["consumesCounters"][0]["counters"] == {
        "memorySlice3": {"value": "1"}
    }
    # counter sets ride on the slice spec
    s = kube.list_resource_slices(DRIVER_NAME)[0]
    names = [c["name"] for c in s["spec"]["sharedCounters"]]
    assert "gpu-0-counters" in names and len(names) == 8


def test_allocating_prospective_partition_auto_carves(driver):
    d, kube, lib = driver
    kube.put_resource_claim(_claim("u-p5", "gpu-2-cpx-5"))
    res = d.node_prepare_resources(
        [type("R", (), {"namespace": "d", "name": "c-u-p5", "uid": "u-p5"})()]
    )
    assert res["u-p5"].error == "", res["u-p5"].error
    assert res["u-p5"].devices[0]["device_name"] == "gpu-2-cpx-5"
    # the GPU was carved by prepare
    g2 = lib.enumerate()[2]
    assert g2.compute_partition == "CPX"
    # publication switched to the REAL partitions for gpu-2 (no longer
    # prospective), still counter-accounted
    devs = _devices(kube)
    assert devs["gpu-2-cpx-5"]["attributes"].get(
        "gpu.amd.com/prospective", {"bool": False}
    )["bool"] is False
    assert "gpu-2" not in devs  # whole-GPU device gone while carved
    # other GPUs keep their prospective devices
    assert devs["gpu-0-cpx-0"]["attributes"]["gpu.amd.com/prospective"]["bool"]


def test_last_pod_drain_restores_spx(driver):
    d, kube, lib = driver
    for uid, dev in (("u-a", "gpu-3-cpx-0"), ("u-b", "gpu-3-cpx-1")):
        kube.put_resource_claim(_claim(uid, dev))
        res = d.node_prepare_resources(
            [type("R", (), {"namespace": "d", "name": f"c-{uid}", "uid": uid})()]
        )
        assert res[uid].error == ""
    assert lib.enumerate()[3].compute_partition == "CPX"
    d.node_unprepare_resources(
        [type("R", (), {"namespace": "d", "name": "c-u-a", "uid": "u-a"})()]
    )
    # still held by u-b: stays carved (drain guard)
    assert lib.enumerate()[3].compute_partition == "CPX"
    d.node_unprepare_resources(
        [type("R", (), {"namespace": "d", "name": "c-u-b", "uid": "u-b"})()]
    )
    # last holder gone: deferred restore returns the GPU to SPX and the
    # whole-GPU device (plus prospective partitions) is republished
    assert lib.enumerate()[3].compute_partition == "SPX"
    devs = _devices(kube)
    assert "gpu-3" in devs
    assert "gpu-3-cpx-0" in devs  # prospective again


def test_v1beta1_apiserver_drops_prospective(tmp_path):
    """Counters need v1beta2+: a v1beta1-only apiserver gets only the
    whole-GPU devices (no overlap the scheduler can't see)."""
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()  # serves v1beta1 only
    d = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        prospective_partitions="cpx",
    )
    d.startup()
    devs = _devices(kube)
    assert len(devs) == 8
    assert all("consumesCounters" not in v for v in devs.values())
    d.shutdown(unpublish=False)


def test_carve_refused_while_whole_gpu_held(driver):
    d, kube, lib = driver
    kube.put_resource_claim(_claim("u-whole", "gpu-4"))
    res = d.node_prepare_resources(
        [
            type(
                "R", (), {"namespace": "d", "name": "c-u-whole", "uid": "u-whole"}
            )()
        ]
    )
    assert res["u-whole"].error == ""
    kube.put_resource_claim(_claim("u-part", "gpu-4-cpx-0"))
    res = d.node_prepare_resources(
        [type("R", (), {"namespace": "d", "name": "c-u-part", "uid": "u-part"})()]
    )
    assert "carving gpu-4" in res["u-part"].error


class TestCounterAwareAllocator:
    """The in-repo allocator (controller scheduler + bench) honors
    sharedCounters overlap like the real kube-scheduler will."""

    def _devices(self):
        from k8s_dra_driver_amd.hal import FakeDeviceLib
        from k8s_dra_driver_amd.hal.model import (
            gpu_device_with_counters,
            prospective_partition_devices,
        )
        from k8s_dra_driver_amd.partition.catalog import make_profile

        lib = FakeDeviceLib()
        lib.open()
        gpus = lib.enumerate()[:2]
        prof = make_profile("CPX", "NPS4")
        devs = []
        for g in gpus:
            devs.append(gpu_device_with_counters(g))
            devs.extend(prospective_partition_devices(g, prof))
        return devs

    def test_whole_gpu_blocks_its_partitions(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        devs = self._devices()
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    {"name": "p", "deviceClassName": "partition.gpu.amd.com"}
                ]
            }
        }
        res = alloc.allocate(spec, devs, pool="n", in_use={"gpu-0"})
        # gpu-0 consumed all gpu-0 counters: only gpu-1 partitions remain
        assert res[0].device.startswith("gpu-1-cpx-")

    def test_partition_blocks_whole_gpu(self):
        import pytest as _pytest

        from k8s_dra_driver_amd.allocator.structured import (
            AllocationError,
            Allocator,
        )

        devs = self._devices()
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    {"name": "g", "deviceClassName": "gpu.amd.com", "count": 2}
                ]
            }
        }
        # one partition of each GPU in use -> no whole GPU allocatable
        with _pytest.raises(AllocationError):
            alloc.allocate(
                spec, devs, pool="n", in_use={"gpu-0-cpx-3", "gpu-1-cpx-0"}
            )

    def test_one_claim_cannot_take_gpu_and_its_partition(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        devs = self._devices()
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    {"name": "g", "deviceClassName": "gpu.amd.com"},
                    {"name": "p", "deviceClassName": "partition.gpu.amd.com"},
                ]
            }
        }
        res = alloc.allocate(spec, devs, pool="n")
        by_req = {r.request: r.device for r in res}
        # the partition must come from the OTHER die
        gpu_idx = by_req["g"].split("-")[1]
        assert not by_req["p"].startswith(f"gpu-{gpu_idx}-")

    def test_disjoint_partitions_coexist(self):
        from k8s_dra_driver_amd.allocator.structured import Allocator

        devs = self._devices()
        alloc = Allocator()
        spec = {
            "devices": {
                "requests": [
                    {"name": "p", "deviceClassName": "partition.gpu.amd.com"}
                ]
            }
        }
        res = alloc.allocate(spec, devs, pool="n", in_use={"gpu-0-cpx-0"})
        assert res[0].device != "gpu-0-cpx-0"  # sibling or other die is fine


def test_concurrent_batch_drain_eventually_restores(driver):
    """8 partition claims unprepared in one concurrent batch: the carve
    owner's restore may be deferred mid-race, but the deferred-restore
    retry (fired on any later unprepare) must bring the GPU back to SPX
    — and re-prepare works either way because prospective and real
    partition names are identical."""
    d, kube, lib = driver
    uids = []
    for k in range(8):
        uid = f"u-b{k}"
        kube.put_resource_claim(_claim(uid, f"gpu-5-cpx-{k}"))
        uids.append(uid)
    refs = [
        type("R", (), {"namespace": "d", "name": f"c-{u}", "uid": u})()
        for u in uids
    ]
    res = d.node_prepare_resources(refs)
    assert all(res[u].error == "" for u in uids)
    assert lib.enumerate()[5].compute_partition == "CPX"
    d.node_unprepare_resources(refs)  # concurrent batch through the pool
    if lib.enumerate()[5].compute_partition != "SPX":
        # race window: retry fires on the next unprepare of anything
        d.state.unprepare("no-such-claim")
        d.state._retry_deferred_restores("sweep")
    assert lib.enumerate()[5].compute_partition == "SPX"
