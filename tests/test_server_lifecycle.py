"""PluginServer lifecycle edge cases."""

import os

import grpc

from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.kube.client import InMemoryKube
from k8s_dra_driver_amd.plugin.driver import Driver
from k8s_dra_driver_amd.plugin.proto import V1BETA1
from k8s_dra_driver_amd.plugin.server import PluginServer


def make_server(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    driver = Driver(
        lib,
        InMemoryKube(),
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
    )
    driver.startup()
    return PluginServer(driver, plugin_dir=str(tmp_path / "plugin"))


def test_stop_without_start_is_safe(tmp_path):
    make_server(tmp_path).stop()


def test_restart_reuses_socket_path(tmp_path):
    server = make_server(tmp_path)
    server.start()
    sock = server.plugin_sock
    server.stop()
    assert not os.path.exists(sock)
    server.start()  # stale-socket path is unlinked and rebound
    assert os.path.exists(sock)
    # serves after restart
    channel = grpc.insecure_channel(f"unix://{sock}")
    m = V1BETA1
    call = channel.unary_unary(
        f"/{m.service_name}/NodePrepareResources",
        request_serializer=lambda x: x.SerializeToString(),
        response_deserializer=m.NodePrepareResourcesResponse.FromString,
    )
    resp = call(m.NodePrepareResourcesRequest(), timeout=5)
    assert resp.claims == {}
    channel.close()
    server.stop()


def test_double_stop_idempotent(tmp_path):
    server = make_server(tmp_path)
    server.start()
    server.stop()
    server.stop()
