"""Watch/informer tests: in-memory hooks, HTTP streaming watch, and the
controller's prompt allocation on claim arrival."""

import time


from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.controller.manager import ControllerManager
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.kube.http_kube import HttpKube
from k8s_dra_driver_amd.kube.miniapiserver import MiniApiServer
from k8s_dra_driver_amd.kube.resourceslice import ResourceSlicePublisher


def _wait_for(pred, timeout=10.0, what=""):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if pred():
            return
        time.sleep(0.05)
    raise TimeoutError(what)


def test_inmemory_watch_events():
    kube = InMemoryKube()
    events = []
    w = kube.watch_resource_claims(lambda t, o: events.append((t, o["metadata"]["name"])))
    kube.put_resource_claim({"metadata": {"namespace": "d", "name": "c1", "uid": "u"}})
    kube.put_resource_claim({"metadata": {"namespace": "d", "name": "c1", "uid": "u"}})
    assert events == [("ADDED", "c1"), ("MODIFIED", "c1")]
    w.stop()
    kube.put_resource_claim({"metadata": {"namespace": "d", "name": "c2", "uid": "u2"}})
    assert len(events) == 2  # stopped


def test_http_watch_streams_events(tmp_path):
    srv = MiniApiServer().start()
    kc = srv.write_kubeconfig(str(tmp_path / "kc"))
    client = HttpKube(kubeconfig=kc, qps=1000, burst=1000)
    events = []
    w = client.watch_resource_claims(lambda t, o: events.append((t, o["metadata"]["name"])))
    time.sleep(0.3)  # let the stream connect
    srv.store.put_resource_claim(
        {"metadata": {"namespace": "d", "name": "cw", "uid": "uw"}}
    )
    _wait_for(lambda: events, what="watch event over HTTP")
    assert events[0] == ("ADDED", "cw")
    w.stop()
    srv.stop()


def test_controller_allocates_promptly_on_watch():
    """A pending claim is allocated on arrival, not after the poll
    interval (set absurdly high to prove the watch path)."""
    kube = InMemoryKube()
    kube.put_node({"metadata": {"name": "n"}})
    lib = FakeDeviceLib()
    lib.open()
    from k8s_dra_driver_amd.hal.model import AllocatableDevice

    ResourceSlicePublisher(kube, driver_name=DRIVER_NAME, node_name="n").publish(
        [AllocatableDevice.from_gpu(g).to_device() for g in lib.enumerate()]
    )
    mgr = ControllerManager(kube, poll_interval=3600, allocate_claims=True)
    mgr.start()
    try:
        time.sleep(0.2)  # initial pass done; loop now parked on the kick
        kube.put_resource_claim(
            {
                "metadata": {"namespace": "d", "name": "late", "uid": "late"},
                "spec": {
                    "devices": {
                        "requests": [
                            {
                                "name": "gpu",
                                "deviceClassName": "gpu.amd.com",
                                "count": 1,
                            }
                        ]
                    }
                },
            }
        )
        _wait_for(
            lambda: (
                kube.get_resource_claim("d", "late").get("status") or {}
            ).get("allocation"),
            timeout=5,
            what="prompt allocation",
        )
    finally:
        mgr.stop()


def test_http_kube_negotiates_v1beta2(tmp_path):
    """End-to-end version negotiation over the wire: apiserver serves
    v1beta2 preferred -> HttpKube uses the v1beta2 REST base and the
    publisher emits flattened devices through it."""
    srv = MiniApiServer().start()
    srv.store.api_versions = ["v1beta2", "v1beta1"]
    kc = srv.write_kubeconfig(str(tmp_path / "kc"))
    client = HttpKube(kubeconfig=kc, qps=1000, burst=1000)
    assert client.resource_api_versions()[0] == "v1beta2"
    pub = ResourceSlicePublisher(
        client, driver_name=DRIVER_NAME, node_name="vn"
    )
    pub.publish(
        [{"name": "gpu-0", "basic": {"attributes": {}, "capacity": {}}}]
    )
    s = srv.store.list_resource_slices(DRIVER_NAME)[0]
    assert s["apiVersion"] == "resource.k8s.io/v1beta2"
    assert "basic" not in s["spec"]["devices"][0]
    srv.stop()


def test_http_kube_defaults_to_v1beta1(tmp_path):
    srv = MiniApiServer().start()
    kc = srv.write_kubeconfig(str(tmp_path / "kc"))
    client = HttpKube(kubeconfig=kc, qps=1000, burst=1000)
    assert client.resource_api_versions() == ["v1beta1"]
    assert client._rbase.endswith("/v1beta1")
    srv.stop()
