"""ResourceSlice publisher reconcile tests."""

from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.kube.resourceslice import (
    MAX_DEVICES_PER_SLICE,
    ResourceSlicePublisher,
)


def _dev(name):
    return {"name": name, "basic": {"attributes": {}, "capacity": {}}}


def _pub(kube):
    return ResourceSlicePublisher(
        kube, driver_name="gpu.amd.com", node_name="node-a"
    )


def test_publish_creates_slice():
    kube = InMemoryKube()
    pub = _pub(kube)
    pub.publish([_dev("gpu-0"), _dev("gpu-1")])
    slices = kube.list_resource_slices("gpu.amd.com")
    assert len(slices) == 1
    assert slices[0]["spec"]["pool"]["generation"] == 1
    assert len(slices[0]["spec"]["devices"]) == 2


def test_unchanged_publish_is_noop():
    kube = InMemoryKube()
    pub = _pub(kube)
    devs = [_dev("gpu-0")]
    pub.publish(devs)
    rv1 = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]["resourceVersion"]
    pub.publish(devs)
    rv2 = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]["resourceVersion"]
    assert rv1 == rv2


def test_changed_devices_bump_generation():
    kube = InMemoryKube()
    pub = _pub(kube)
    pub.publish([_dev("gpu-0")])
    pub.publish([_dev("gpu-0-cpx-0"), _dev("gpu-0-cpx-1")])
    s = kube.list_resource_slices("gpu.amd.com")[0]
    assert s["spec"]["pool"]["generation"] == 2
    assert len(s["spec"]["devices"]) == 2


def test_chunking_over_max_devices():
    kube = InMemoryKube()
    pub = _pub(kube)
    many = [_dev(f"d-{i}") for i in range(MAX_DEVICES_PER_SLICE + 5)]
    pub.publish(many)
    slices = kube.list_resource_slices("gpu.amd.com")
    assert len(slices) == 2
    assert all(
        s["spec"]["pool"]["resourceSliceCount"] == 2 for s in slices
    )
    total = sum(len(s["spec"]["devices"]) for s in slices)
    assert total == len(many)
    # shrink back to one slice: surplus slice must be deleted
    pub.publish(many[:3])
    assert len(kube.list_resource_slices("gpu.amd.com")) == 1


def test_unpublish_all():
    kube = InMemoryKube()
    pub = _pub(kube)
    pub.publish([_dev("gpu-0")])
    pub.unpublish_all()
    assert kube.list_resource_slices("gpu.amd.com") == []


def test_owner_reference_when_node_uid_known():
    kube = InMemoryKube()
    pub = ResourceSlicePublisher(
        kube, driver_name="gpu.amd.com", node_name="node-a", node_uid="node-uid-1"
    )
    pub.publish([_dev("gpu-0")])
    meta = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]
    assert meta["ownerReferences"][0] == {
        "apiVersion": "v1",
        "kind": "Node",
        "name": "node-a",
        "uid": "node-uid-1",
    }


def test_driver_shutdown_unpublishes(tmp_path):
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.plugin.driver import Driver

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "n", "uid": "nuid"}})
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    slices = kube.list_resource_slices("gpu.amd.com")
    assert slices
    assert slices[0]["metadata"]["ownerReferences"][0]["uid"] == "nuid"
    driver.shutdown()
    assert kube.list_resource_slices("gpu.amd.com") == []


def test_generation_survives_publisher_restart():
    """A restarted plugin must not regress the pool generation."""
    kube = InMemoryKube()
    pub1 = _pub(kube)
    pub1.publish([_dev("gpu-0")])
    pub1.publish([_dev("gpu-0"), _dev("gpu-1")])  # generation 2
    pub2 = _pub(kube)  # restart
    pub2.publish([_dev("gpu-0")])  # changed set -> must bump past 2
    gen = kube.list_resource_slices("gpu.amd.com")[0]["spec"]["pool"][
        "generation"
    ]
    assert gen == 3


class TestDriftSelfHeal:
    """Kill-a-slice recovery without a repartition event (VERDICT r1 #2;
    reference resourceslicecontroller.go:407-431)."""

    def test_external_delete_recovers(self):
        kube = InMemoryKube()
        pub = _pub(kube)
        pub.publish([_dev("gpu-0"), _dev("gpu-1")])
        assert pub.start_self_heal()
        name = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]["name"]
        kube.delete_resource_slice(name)  # tampering, synchronous hook
        slices = kube.list_resource_slices("gpu.amd.com")
        assert len(slices) == 1
        assert [d["name"] for d in slices[0]["spec"]["devices"]] == [
            "gpu-0",
            "gpu-1",
        ]
        assert pub.heal_count == 1
        # healed publication bumped the pool generation (scheduler must
        # treat the re-published slice as authoritative)
        assert slices[0]["spec"]["pool"]["generation"] == 2
        pub.stop_self_heal()

    def test_external_mutation_recovers(self):
        kube = InMemoryKube()
        pub = _pub(kube)
        pub.publish([_dev("gpu-0"), _dev("gpu-1")])
        assert pub.start_self_heal()
        cur = kube.list_resource_slices("gpu.amd.com")[0]
        cur["spec"]["devices"] = [_dev("rogue")]
        kube.update_resource_slice(cur)  # tampering
        slices = kube.list_resource_slices("gpu.amd.com")
        assert [d["name"] for d in slices[0]["spec"]["devices"]] == [
            "gpu-0",
            "gpu-1",
        ]
        assert pub.heal_count >= 1
        pub.stop_self_heal()

    def test_own_updates_do_not_self_trigger(self):
        kube = InMemoryKube()
        pub = _pub(kube)
        pub.publish([_dev("gpu-0")])
        assert pub.start_self_heal()
        pub.publish([_dev("gpu-0"), _dev("gpu-1")])  # normal repartition
        assert pub.heal_count == 0

    def test_publish_detects_drift_without_watch(self):
        """Even pollers recover: the fingerprint short-circuit is bypassed
        when observed slices diverge from desired."""
        kube = InMemoryKube()
        pub = _pub(kube)
        devs = [_dev("gpu-0")]
        pub.publish(devs)
        name = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]["name"]
        kube.delete_resource_slice(name)
        pub.publish(devs)  # same fingerprint, but observed state drifted
        assert len(kube.list_resource_slices("gpu.amd.com")) == 1


class TestV1beta2Publication:
    """Version-negotiated publication: one code path,
    flattened Device shape when the apiserver serves v1beta2."""

    def _basic_dev(self):
        return {
            "name": "gpu-0",
            "basic": {
                "attributes": {"gpu.amd.com/type": {"string": "gpu"}},
                "capacity": {"gpu.amd.com/memory": {"value": "288Gi"}},
            },
        }

    def test_v1beta1_default_keeps_basic_shape(self):
        kube = InMemoryKube()
        pub = _pub(kube)
        pub.publish([self._basic_dev()])
        s = kube.list_resource_slices("gpu.amd.com")[0]
        assert s["apiVersion"] == "resource.k8s.io/v1beta1"
        assert "basic" in s["spec"]["devices"][0]

    def test_v1beta2_flattens_devices(self):
        kube = InMemoryKube()
        kube.api_versions = ["v1beta2", "v1beta1"]
        pub = _pub(kube)
        pub.publish([self._basic_dev()])
        s = kube.list_resource_slices("gpu.amd.com")[0]
        assert s["apiVersion"] == "resource.k8s.io/v1beta2"
        d = s["spec"]["devices"][0]
        assert "basic" not in d
        assert d["attributes"]["gpu.amd.com/type"] == {"string": "gpu"}
        assert d["capacity"]["gpu.amd.com/memory"] == {"value": "288Gi"}

    def test_v1beta2_self_heal_keeps_flat_shape(self):
        kube = InMemoryKube()
        kube.api_versions = ["v1beta2"]
        pub = _pub(kube)
        pub.publish([self._basic_dev()])
        assert pub.start_self_heal()
        name = kube.list_resource_slices("gpu.amd.com")[0]["metadata"]["name"]
        kube.delete_resource_slice(name)
        s = kube.list_resource_slices("gpu.amd.com")[0]
        assert s["apiVersion"] == "resource.k8s.io/v1beta2"
        assert "basic" not in s["spec"]["devices"][0]
        pub.stop_self_heal()

    def test_v1_ga_flattens_devices(self):
        kube = InMemoryKube()
        kube.api_versions = ["v1", "v1beta1"]
        pub = _pub(kube)
        pub.publish([self._basic_dev()])
        s = kube.list_resource_slices("gpu.amd.com")[0]
        assert s["apiVersion"] == "resource.k8s.io/v1"
        assert "basic" not in s["spec"]["devices"][0]
