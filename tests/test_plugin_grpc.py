"""End-to-end plugin tests: real gRPC over unix sockets, fake kubelet client,
in-memory apiserver, fake HAL. This is BASELINE config #1's plumbing path.
"""

import os

import grpc
import pytest

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import Driver
from k8s_dra_driver_amd.plugin.proto import REGISTRATION, V1ALPHA4, V1BETA1
from k8s_dra_driver_amd.plugin.server import PluginServer

NODE = "node-a"


@pytest.fixture
def stack(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name=NODE,
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "ckpt"),
        use_tmpfs=False,
    )
    driver.startup()
    server = PluginServer(
        driver,
        plugin_dir=str(tmp_path / "plugins" / DRIVER_NAME),
        registry_dir=str(tmp_path / "plugins_registry"),
    )
    server.start()
    channel = grpc.insecure_channel(f"unix://{server.plugin_sock}")
    yield lib, kube, driver, server, channel
    channel.close()
    server.stop()


def _stub(channel, msgs, method, req_cls, resp_cls):
    return channel.unary_unary(
        f"/{msgs.service_name}/{method}",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=resp_cls.FromString,
    )


def put_claim(kube, uid, devices, configs=None):
    claim = {
        "metadata": {"namespace": "default", "name": f"claim-{uid}", "uid": uid},
        "status": {
            "allocation": {
                "devices": {
                    "results": [
                        {
                            "request": "gpu",
                            "driver": DRIVER_NAME,
                            "pool": NODE,
                            "device": d,
                        }
                        for d in devices
                    ],
                    "config": configs or [],
                }
            }
        },
    }
    kube.put_resource_claim(claim)
    return claim


def test_registration_get_info(stack, tmp_path):
    lib, kube, driver, server, channel = stack
    reg_channel = grpc.insecure_channel(f"unix://{server.registry_sock}")
    get_info = reg_channel.unary_unary(
        f"/{REGISTRATION.service_name}/GetInfo",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=REGISTRATION.PluginInfo.FromString,
    )
    info = get_info(REGISTRATION.InfoRequest(), timeout=5)
    assert info.type == "DRAPlugin"
    assert info.name == DRIVER_NAME
    assert info.endpoint == server.plugin_sock
    assert list(info.supported_versions) == ["v1", "v1beta1", "v1alpha4"]
    reg_channel.close()


def test_resourceslices_published_at_startup(stack):
    _, kube, *_ = stack
    slices = kube.list_resource_slices(DRIVER_NAME)
    assert len(slices) == 1
    spec = slices[0]["spec"]
    assert spec["nodeName"] == NODE
    assert len(spec["devices"]) == 8
    names = [d["name"] for d in spec["devices"]]
    assert names == [f"gpu-{i}" for i in range(8)]


@pytest.mark.parametrize("msgs", [V1BETA1, V1ALPHA4], ids=["v1beta1", "v1alpha4"])
def test_prepare_unprepare_roundtrip(stack, msgs):
    lib, kube, driver, server, channel = stack
    put_claim(kube, "uid-1", ["gpu-0"])
    prepare = _stub(
        channel,
        msgs,
        "NodePrepareResources",
        msgs.NodePrepareResourcesRequest,
        msgs.NodePrepareResourcesResponse,
    )
    req = msgs.NodePrepareResourcesRequest()
    c = req.claims.add()
    c.namespace, c.name, c.uid = "default", "claim-uid-1", "uid-1"
    resp = prepare(req, timeout=10)
    assert resp.claims["uid-1"].error == ""
    devs = resp.claims["uid-1"].devices
    assert len(devs) == 1
    assert devs[0].device_name == "gpu-0"
    assert devs[0].pool_name == NODE
    assert list(devs[0].cdi_device_ids) == [
        "k8s.gpu.amd.com/device=gpu-0",
        "k8s.gpu.amd.com/claim=uid-1-gpu-0",
    ]

    unprepare = _stub(
        channel,
        msgs,
        "NodeUnprepareResources",
        msgs.NodeUnprepareResourcesRequest,
        msgs.NodeUnprepareResourcesResponse,
    )
    ureq = msgs.NodeUnprepareResourcesRequest()
    uc = ureq.claims.add()
    uc.namespace, uc.name, uc.uid = "default", "claim-uid-1", "uid-1"
    uresp = unprepare(ureq, timeout=10)
    assert uresp.claims["uid-1"].error == ""
    assert driver.state.checkpoints.read("uid-1") is None


def test_per_claim_errors_in_response_map(stack):
    lib, kube, driver, server, channel = stack
    put_claim(kube, "uid-ok", ["gpu-1"])
    # uid-missing has no ResourceClaim object; uid-bad references a bogus device
    put_claim(kube, "uid-bad", ["gpu-99"])
    msgs = V1BETA1
    prepare = _stub(
        channel,
        msgs,
        "NodePrepareResources",
        msgs.NodePrepareResourcesRequest,
        msgs.NodePrepareResourcesResponse,
    )
    req = msgs.NodePrepareResourcesRequest()
    for ns, name, uid in [
        ("default", "claim-uid-ok", "uid-ok"),
        ("default", "claim-uid-missing", "uid-missing"),
        ("default", "claim-uid-bad", "uid-bad"),
    ]:
        c = req.claims.add()
        c.namespace, c.name, c.uid = ns, name, uid
    resp = prepare(req, timeout=10)
    assert resp.claims["uid-ok"].error == ""
    assert "not found" in resp.claims["uid-missing"].error
    assert "not found" in resp.claims["uid-bad"].error
    # the good claim really was prepared despite sibling failures
    assert driver.state.checkpoints.read("uid-ok") is not None


def test_uid_mismatch_rejected(stack):
    lib, kube, driver, server, channel = stack
    put_claim(kube, "uid-new", ["gpu-2"])
    msgs = V1BETA1
    prepare = _stub(
        channel,
        msgs,
        "NodePrepareResources",
        msgs.NodePrepareResourcesRequest,
        msgs.NodePrepareResourcesResponse,
    )
    req = msgs.NodePrepareResourcesRequest()
    c = req.claims.add()
    # kubelet asks for an old incarnation of the claim
    c.namespace, c.name, c.uid = "default", "claim-uid-new", "uid-old"
    resp = prepare(req, timeout=10)
    assert "UID mismatch" in resp.claims["uid-old"].error


def test_repartition_republishes_slices(stack):
    lib, kube, driver, server, channel = stack
    cfg = {
        "source": "FromClaim",
        "requests": [],
        "opaque": {
            "driver": DRIVER_NAME,
            "parameters": {
                "apiVersion": API_GROUP_VERSION,
                "kind": "PartitionConfig",
                "computePartition": "CPX",
                "memoryPartition": "NPS4",
                "allowDynamicRepartition": True,
            },
        },
    }
    put_claim(kube, "uid-part", ["gpu-0"], configs=[cfg])
    msgs = V1BETA1
    prepare = _stub(
        channel,
        msgs,
        "NodePrepareResources",
        msgs.NodePrepareResourcesRequest,
        msgs.NodePrepareResourcesResponse,
    )
    req = msgs.NodePrepareResourcesRequest()
    c = req.claims.add()
    c.namespace, c.name, c.uid = "default", "claim-uid-part", "uid-part"
    resp = prepare(req, timeout=10)
    assert resp.claims["uid-part"].error == ""
    assert len(resp.claims["uid-part"].devices) == 8  # whole carved die
    # ResourceSlices now advertise the partitions with a bumped generation
    slices = kube.list_resource_slices(DRIVER_NAME)
    names = [d["name"] for d in slices[0]["spec"]["devices"]]
    assert "gpu-0-cpx-0" in names and "gpu-1" in names
    assert slices[0]["spec"]["pool"]["generation"] >= 2


def test_batch_prepare_is_concurrent(stack):
    """8 claims in one batch must not serialize behind one lock."""
    lib, kube, driver, server, channel = stack
    lib.faults.set_latency("device_node_paths", 0.0)  # no-op guard
    for i in range(8):
        put_claim(kube, f"uid-b{i}", [f"gpu-{i}"])
    msgs = V1BETA1
    prepare = _stub(
        channel,
        msgs,
        "NodePrepareResources",
        msgs.NodePrepareResourcesRequest,
        msgs.NodePrepareResourcesResponse,
    )
    req = msgs.NodePrepareResourcesRequest()
    for i in range(8):
        c = req.claims.add()
        c.namespace, c.name, c.uid = "default", f"claim-uid-b{i}", f"uid-b{i}"
    resp = prepare(req, timeout=20)
    assert all(resp.claims[f"uid-b{i}"].error == "" for i in range(8))


def test_malformed_request_bytes_do_not_crash_server(stack):
    """Garbage on the wire must yield a gRPC error, not kill the plugin."""
    lib, kube, driver, server, channel = stack
    m = V1BETA1
    raw = channel.unary_unary(
        f"/{m.service_name}/NodePrepareResources",
        request_serializer=lambda b: b,  # send raw bytes
        response_deserializer=lambda b: b,
    )
    with pytest.raises(grpc.RpcError):
        raw(b"\xff\xfe\xfd this is not protobuf \x00\x01", timeout=5)
    # server still alive and serving
    put_claim(kube, "uid-after", ["gpu-4"])
    prepare = _stub(
        channel,
        m,
        "NodePrepareResources",
        m.NodePrepareResourcesRequest,
        m.NodePrepareResourcesResponse,
    )
    req = m.NodePrepareResourcesRequest()
    c = req.claims.add()
    c.namespace, c.name, c.uid = "default", "claim-uid-after", "uid-after"
    assert prepare(req, timeout=10).claims["uid-after"].error == ""


def test_socket_watchdog_rebinds(stack):
    """kubelet-restart resilience: a wiped socket is re-bound."""
    import os
    import time

    lib, kube, driver, server, channel = stack
    server.start_socket_watchdog(interval_s=0.2)
    os.unlink(server.registry_sock)
    t0 = time.time()
    while not os.path.exists(server.registry_sock) and time.time() - t0 < 10:
        time.sleep(0.1)
    assert os.path.exists(server.registry_sock), "socket not re-bound"
    # and it serves again
    reg_channel = grpc.insecure_channel(f"unix://{server.registry_sock}")
    info = reg_channel.unary_unary(
        f"/{REGISTRATION.service_name}/GetInfo",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=REGISTRATION.PluginInfo.FromString,
    )(REGISTRATION.InfoRequest(), timeout=5)
    assert info.name == DRIVER_NAME
    reg_channel.close()


def test_v1_service_prepares(stack):
    """K8s 1.34 kubelets speak the GA dra.v1 service; same wire shapes."""
    import grpc as _grpc

    from k8s_dra_driver_amd.plugin.proto import V1

    lib, kube, driver, server, _channel = stack
    kube.put_resource_claim(
        {
            "metadata": {"namespace": "d", "name": "cv1", "uid": "uid-v1"},
            "status": {
                "allocation": {
                    "devices": {
                        "results": [
                            {
                                "request": "gpu",
                                "driver": DRIVER_NAME,
                                "pool": NODE,
                                "device": "gpu-1",
                            }
                        ]
                    }
                }
            },
        }
    )
    channel = _grpc.insecure_channel(f"unix://{server.plugin_sock}")
    m = V1
    assert m.service_name == "k8s.io.kubelet.pkg.apis.dra.v1.DRAPlugin"
    prepare = channel.unary_unary(
        f"/{m.service_name}/NodePrepareResources",
        request_serializer=lambda x: x.SerializeToString(),
        response_deserializer=m.NodePrepareResourcesResponse.FromString,
    )
    req = m.NodePrepareResourcesRequest()
    c = req.claims.add()
    c.namespace, c.name, c.uid = "d", "cv1", "uid-v1"
    resp = prepare(req, timeout=10)
    assert resp.claims["uid-v1"].error == ""
    assert resp.claims["uid-v1"].devices[0].device_name == "gpu-1"
    channel.close()
