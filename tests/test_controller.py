"""Controller tests: node labeling from slices, orphan cleanup."""

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.controller.manager import ControllerManager, labels_for_node
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.kube.resourceslice import ResourceSlicePublisher


def publish_node(kube, node, lib):
    from k8s_dra_driver_amd.hal.model import AllocatableDevice

    devs = []
    for g in lib.enumerate():
        if g.partitions:
            devs.extend(
                AllocatableDevice.from_partition(g, p).to_device()
                for p in g.partitions
            )
        else:
            devs.append(AllocatableDevice.from_gpu(g).to_device())
    ResourceSlicePublisher(kube, driver_name=DRIVER_NAME, node_name=node).publish(
        devs
    )


def test_labels_for_node(fake_lib):
    from k8s_dra_driver_amd.hal.model import AllocatableDevice

    devs = [
        AllocatableDevice.from_gpu(g).to_device() for g in fake_lib.enumerate()
    ]
    labels = labels_for_node(devs)
    assert labels["gpu.amd.com/gpu.present"] == "true"
    assert labels["gpu.amd.com/gpu.count"] == "8"
    assert labels["gpu.amd.com/gpu.architecture"] == "gfx950"
    assert labels["gpu.amd.com/xgmi.hive"].startswith("hive-")
    assert labels["gpu.amd.com/partition.modes"] == "SPX"


def test_reconcile_labels_nodes():
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "node-a"}})
    lib = FakeDeviceLib()
    lib.open()
    publish_node(kube, "node-a", lib)
    mgr = ControllerManager(kube)
    applied = mgr.reconcile_once()
    assert "node-a" in applied
    node = kube.get_node("node-a")
    assert node["metadata"]["labels"]["gpu.amd.com/gpu.count"] == "8"


def test_partition_count_still_counts_gpus():
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "node-a"}})
    lib = FakeDeviceLib()
    lib.open()
    lib.set_compute_partition(0, "CPX")
    publish_node(kube, "node-a", lib)
    ControllerManager(kube).reconcile_once()
    labels = kube.get_node("node-a")["metadata"]["labels"]
    assert labels["gpu.amd.com/gpu.count"] == "8"  # 7 whole + 1 carved
    assert labels["gpu.amd.com/device.count"] == "15"
    assert "CPX" in labels["gpu.amd.com/partition.modes"]


def test_orphan_slice_cleanup():
    kube = InMemoryKube()
    lib = FakeDeviceLib()
    lib.open()
    publish_node(kube, "gone-node", lib)  # node object never created
    assert kube.list_resource_slices(DRIVER_NAME)
    ControllerManager(kube).reconcile_once()
    assert kube.list_resource_slices(DRIVER_NAME) == []


def test_labels_removed_when_slices_gone():
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "node-a"}})
    lib = FakeDeviceLib()
    lib.open()
    pub = ResourceSlicePublisher(
        kube, driver_name=DRIVER_NAME, node_name="node-a"
    )
    publish_node(kube, "node-a", lib)
    mgr = ControllerManager(kube)
    mgr.reconcile_once()
    assert "gpu.amd.com/gpu.count" in kube.get_node("node-a")["metadata"]["labels"]
    pub.unpublish_all()
    mgr.reconcile_once()
    labels = kube.get_node("node-a")["metadata"]["labels"]
    assert "gpu.amd.com/gpu.count" not in labels


def test_gpu_count_with_prospective_publication(tmp_path):
    """gpu.count must count DIES, not devices: prospective-partition mode
    publishes each whole GPU alongside its would-be partitions."""
    from k8s_dra_driver_amd import DRIVER_NAME
    from k8s_dra_driver_amd.controller.manager import labels_for_node
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    kube.api_versions = ["v1beta2", "v1beta1"]
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
    devices = [
        dev
        for s in kube.list_resource_slices(DRIVER_NAME)
        for dev in s["spec"]["devices"]
    ]
    assert len(devices) == 72  # 8 whole + 64 prospective
    labels = labels_for_node(devices)
    assert labels["gpu.amd.com/gpu.count"] == "8"
    assert labels["gpu.amd.com/device.count"] == "72"
    d.shutdown(unpublish=False)
