"""Health monitor tests: fault-injected failure detection + slice healing."""

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.base import HalError
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import Driver
from k8s_dra_driver_amd.plugin.health import HealthMonitor


def test_hysteresis_and_recovery(fake_lib):
    events = []
    mon = HealthMonitor(
        fake_lib,
        on_change=lambda s: events.append(set(s)),
        failures_to_unhealthy=2,
    )
    # one failure: below threshold
    fake_lib.faults.fail_next("health_check", HalError("hw error"))
    assert mon.check_once() == set()
    # second consecutive failure on gpu-0 -> unhealthy
    fake_lib.faults.fail_next("health_check", HalError("hw error"))
    assert mon.check_once() == {0}
    assert events == [{0}]
    # healthy pass -> recovered
    assert mon.check_once() == set()
    assert events == [{0}, set()]


def test_unhealthy_gpu_pulled_from_slices(tmp_path):
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
    assert (
        len(kube.list_resource_slices(DRIVER_NAME)[0]["spec"]["devices"]) == 8
    )
    # gpu-0 fails in two consecutive polls (the injector pops one error
    # per call; the first health_check of each round is gpu-0)
    driver.health.failures_to_unhealthy = 2
    lib.faults.fail_next("health_check", HalError("dead"))
    driver.health.check_once()
    lib.faults.fail_next("health_check", HalError("dead"))
    driver.health.check_once()
    devices = kube.list_resource_slices(DRIVER_NAME)[0]["spec"]["devices"]
    assert len(devices) == 7
    assert "gpu-0" not in [d["name"] for d in devices]


def test_device_kinds_gating(tmp_path):
    """--device-classes parity: publish only the enabled kinds."""
    lib = FakeDeviceLib()
    lib.open()
    lib.set_compute_partition(0, "CPX")
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        device_kinds=["partition"],
    )
    driver.startup()
    names = [
        d["name"]
        for s in kube.list_resource_slices(DRIVER_NAME)
        for d in s["spec"]["devices"]
    ]
    assert names and all("cpx" in n for n in names)


class TestUnhealthyDeviceTaints:
    """DRA device taints (K8s 1.33): on v1beta2+ unhealthy GPUs stay
    published but tainted NoSchedule; on v1beta1 they are removed."""

    def _driver(self, tmp_path, api_versions):
        from k8s_dra_driver_amd.hal import FakeDeviceLib
        from k8s_dra_driver_amd.kube.client import InMemoryKube
        from k8s_dra_driver_amd.plugin.driver import Driver

        lib = FakeDeviceLib()
        lib.open()
        kube = InMemoryKube()
        kube.api_versions = api_versions
        d = Driver(
            lib,
            kube,
            node_name="n",
            cdi_root=str(tmp_path / "cdi"),
            checkpoint_root=str(tmp_path / "state"),
            use_tmpfs=False,
        )
        d.startup()
        return d, kube, lib

    def _names(self, kube):
        from k8s_dra_driver_amd import DRIVER_NAME

        return {
            dev["name"]: dev
            for s in kube.list_resource_slices(DRIVER_NAME)
            for dev in s["spec"]["devices"]
        }

    def test_v1beta2_taints_instead_of_removal(self, tmp_path):
        d, kube, lib = self._driver(tmp_path, ["v1beta2", "v1beta1"])
        with d.health._lock:
            d.health._unhealthy = {2}
        d.publish_resources()
        devs = self._names(kube)
        assert "gpu-2" in devs  # still published
        assert devs["gpu-2"]["taints"] == [
            {"key": "gpu.amd.com/unhealthy", "effect": "NoSchedule"}
        ]
        assert "taints" not in devs["gpu-1"]
        # recovery clears the taint
        with d.health._lock:
            d.health._unhealthy = set()
        d.publish_resources()
        assert "taints" not in self._names(kube)["gpu-2"]
        d.shutdown(unpublish=False)

    def test_v1beta1_removes_as_before(self, tmp_path):
        d, kube, lib = self._driver(tmp_path, ["v1beta1"])
        with d.health._lock:
            d.health._unhealthy = {5}
        d.publish_resources()
        assert "gpu-5" not in self._names(kube)
        d.shutdown(unpublish=False)
