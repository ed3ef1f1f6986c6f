"""Controller claim-scheduler tests: the topology-aware allocation loop."""

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.controller.manager import ControllerManager
from k8s_dra_driver_amd.controller.scheduler import ClaimScheduler
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.model import AllocatableDevice
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.kube.resourceslice import ResourceSlicePublisher


def publish(kube, node, lib):
    devs = []
    for g in lib.enumerate():
        if g.partitions:
            devs.extend(
                AllocatableDevice.from_partition(g, p).to_device()
                for p in g.partitions
            )
        else:
            devs.append(AllocatableDevice.from_gpu(g).to_device())
    ResourceSlicePublisher(kube, driver_name=DRIVER_NAME, node_name=node).publish(devs)


def pending_claim(kube, uid, count=1, cls="gpu.amd.com", constraints=None):
    claim = {
        "metadata": {"namespace": "d", "name": f"c-{uid}", "uid": uid},
        "spec": {
            "devices": {
                "requests": [
                    {"name": "gpu", "deviceClassName": cls, "count": count}
                ],
                **({"constraints": constraints} if constraints else {}),
            }
        },
    }
    kube.put_resource_claim(claim)
    return claim


def test_allocates_pending_claims():
    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    publish(kube, "node-a", lib)
    pending_claim(kube, "u1")
    pending_claim(kube, "u2", count=4)
    sched = ClaimScheduler(kube)
    assert sorted(sched.reconcile_once()) == ["u1", "u2"]
    c1 = kube.get_resource_claim("d", "c-u1")
    assert c1["status"]["allocation"]["devices"]["results"]
    # second pass: nothing left to do
    assert sched.reconcile_once() == []
    # no device allocated twice
    allocated = [
        r["device"]
        for c in kube.list_resource_claims()
        for r in c["status"]["allocation"]["devices"]["results"]
    ]
    assert len(allocated) == len(set(allocated)) == 5


def test_unsatisfiable_claim_left_pending():
    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    publish(kube, "node-a", lib)
    pending_claim(kube, "big", count=9)  # only 8 GPUs
    assert ClaimScheduler(kube).reconcile_once() == []
    assert "allocation" not in (
        kube.get_resource_claim("d", "c-big").get("status") or {}
    )


def test_foreign_class_ignored():
    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    publish(kube, "node-a", lib)
    pending_claim(kube, "other", cls="gpu.nvidia.com")
    assert ClaimScheduler(kube).reconcile_once() == []


def test_multi_node_picks_feasible_node():
    kube = InMemoryKube()
    libA = FakeDeviceLib()
    libA.open()
    publish(kube, "node-a", libA)
    from k8s_dra_driver_amd.hal import FakeNodeConfig

    libB = FakeDeviceLib(FakeNodeConfig(num_gpus=2, hive_id="hive-b"))
    libB.open()
    publish(kube, "node-b", libB)
    # 4-GPU claim fits only node-a (node-b has 2)
    pending_claim(kube, "quad", count=4)
    sched = ClaimScheduler(kube)
    assert sched.reconcile_once() == ["quad"]
    c = kube.get_resource_claim("d", "c-quad")
    terms = c["status"]["allocation"]["nodeSelector"]["nodeSelectorTerms"]
    assert terms[0]["matchFields"][0]["values"] == ["node-a"]


def test_cluster_device_classes_honored():
    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    publish(kube, "node-a", lib)
    kube.put_device_class(
        {
            "metadata": {"name": "evengpu.amd.com"},
            "spec": {
                "selectors": [
                    {
                        "cel": {
                            "expression": "device.attributes['gpu.amd.com'].index in [0, 2, 4, 6]"
                        }
                    }
                ]
            },
        }
    )
    pending_claim(kube, "even", cls="evengpu.amd.com", count=2)
    assert ClaimScheduler(kube).reconcile_once() == ["even"]
    devices = [
        r["device"]
        for r in kube.get_resource_claim("d", "c-even")["status"]["allocation"][
            "devices"
        ]["results"]
    ]
    assert all(int(d.split("-")[1]) % 2 == 0 for d in devices)


def test_manager_integration():
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "node-a"}})
    lib = FakeDeviceLib()
    lib.open()
    publish(kube, "node-a", lib)
    pending_claim(kube, "u1")
    mgr = ControllerManager(kube, allocate_claims=True)
    mgr.reconcile_once()
    assert kube.get_resource_claim("d", "c-u1")["status"].get("allocation")


def test_cross_node_topology_placement(tmp_path):
    """Two simulated nodes (one plugin instance each via --gpu-indices
    scoping, nvkind analog): a 3-GPU topology claim must land WHOLLY on
    one node — the controller scores nodes and never splits a claim."""
    from k8s_dra_driver_amd.hal import FakeDeviceLib, FakeNodeConfig
    from k8s_dra_driver_amd.plugin.driver import Driver

    kube = InMemoryKube()
    drivers = []
    # node-a manages 2 GPUs, node-b manages 4: only node-b fits count=3
    for node, idxs in (("node-a", [0, 1]), ("node-b", [2, 3, 4, 5])):
        lib = FakeDeviceLib(FakeNodeConfig(num_gpus=8))
        lib.open()
        d = Driver(
            lib,
            kube,
            node_name=node,
            cdi_root=str(tmp_path / node / "cdi"),
            checkpoint_root=str(tmp_path / node / "state"),
            use_tmpfs=False,
            gpu_indices=idxs,
        )
        d.startup()
        drivers.append(d)

    claim = {
        "metadata": {"namespace": "d", "name": "tri", "uid": "uid-tri"},
        "spec": {
            "devices": {
                "requests": [
                    {
                        "name": "tri",
                        "deviceClassName": "gpu.amd.com",
                        "count": 3,
                    }
                ]
            }
        },
    }
    kube.put_resource_claim(claim)
    sched = ClaimScheduler(kube)
    assert sched.reconcile_once() == ["uid-tri"]
    stored = kube.get_resource_claim("d", "tri")
    alloc = stored["status"]["allocation"]
    results = alloc["devices"]["results"]
    assert len(results) == 3
    pools = {r["pool"] for r in results}
    assert pools == {"node-b"}  # whole claim on the only node that fits
    # nodeSelector pins the pod to that node
    terms = alloc["nodeSelector"]["nodeSelectorTerms"]
    assert any(
        any(
            "node-b" in (m.get("values") or [])
            for m in t.get("matchFields", []) + t.get("matchExpressions", [])
        )
        for t in terms
    )
    for d in drivers:
        d.shutdown(unpublish=False)
