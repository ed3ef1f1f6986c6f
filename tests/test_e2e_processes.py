"""Multi-process e2e: the REAL CLI binaries (kubelet plugin + controller)
as subprocesses against the mini apiserver, driven by a fake kubelet over
gRPC. Validates CLI wiring, HttpKube auth, registration sockets, slice
publication, claim prepare, controller labeling — the closest thing to a
kind cluster that runs in CI.
"""

import os
import subprocess
import sys
import time

import grpc
import pytest

from k8s_dra_driver_amd import DRIVER_NAME
from k8s_dra_driver_amd.kube.miniapiserver import MiniApiServer
from k8s_dra_driver_amd.plugin.proto import REGISTRATION, V1BETA1

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _wait_for(predicate, timeout=20.0, what=""):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if predicate():
            return
        time.sleep(0.1)
    raise TimeoutError(f"timed out waiting for {what}")


@pytest.fixture
def api(tmp_path):
    srv = MiniApiServer().start()
    srv.store.put_node({"metadata": {"name": "e2e-node"}})
    kubeconfig = srv.write_kubeconfig(str(tmp_path / "kubeconfig"))
    yield srv, kubeconfig
    srv.stop()


@pytest.mark.timeout(120)
def test_plugin_process_end_to_end(api, tmp_path):
    srv, kubeconfig = api
    plugin_dir = tmp_path / "plugins" / DRIVER_NAME
    registry_dir = tmp_path / "plugins_registry"
    env = dict(
        os.environ,
        PYTHONPATH=REPO,
        NODE_NAME="e2e-node",
        KUBECONFIG=kubeconfig,
    )
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "k8s_dra_driver_amd.plugin.main",
            "--hal",
            "fake",
            "--cdi-root",
            str(tmp_path / "cdi"),
            "--plugin-path",
            str(plugin_dir),
            "--plugin-registration-path",
            str(registry_dir),
        ],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        reg_sock = registry_dir / f"{DRIVER_NAME}.sock"
        _wait_for(reg_sock.exists, what="registration socket")
        _wait_for(
            lambda: srv.store.list_resource_slices(DRIVER_NAME),
            what="ResourceSlice publication",
        )
        slices = srv.store.list_resource_slices(DRIVER_NAME)
        assert len(slices[0]["spec"]["devices"]) == 8

        # kubelet handshake
        reg_channel = grpc.insecure_channel(f"unix://{reg_sock}")
        info = reg_channel.unary_unary(
            f"/{REGISTRATION.service_name}/GetInfo",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=REGISTRATION.PluginInfo.FromString,
        )(REGISTRATION.InfoRequest(), timeout=10)
        assert info.name == DRIVER_NAME
        reg_channel.close()

        # schedule a claim (controller-side allocation done inline here)
        srv.store.put_resource_claim(
            {
                "metadata": {"namespace": "d", "name": "c1", "uid": "e2e-uid"},
                "status": {
                    "allocation": {
                        "devices": {
                            "results": [
                                {
                                    "request": "gpu",
                                    "driver": DRIVER_NAME,
                                    "pool": "e2e-node",
                                    "device": "gpu-0",
                                }
                            ]
                        }
                    }
                },
            }
        )
        m = V1BETA1
        channel = grpc.insecure_channel(f"unix://{info.endpoint}")
        prepare = channel.unary_unary(
            f"/{m.service_name}/NodePrepareResources",
            request_serializer=lambda x: x.SerializeToString(),
            response_deserializer=m.NodePrepareResourcesResponse.FromString,
        )
        req = m.NodePrepareResourcesRequest()
        c = req.claims.add()
        c.namespace, c.name, c.uid = "d", "c1", "e2e-uid"
        resp = prepare(req, timeout=15)
        assert resp.claims["e2e-uid"].error == ""
        assert resp.claims["e2e-uid"].devices[0].device_name == "gpu-0"
        # CDI claim spec written by the plugin process
        assert (tmp_path / "cdi").glob("*claim*")
        channel.close()
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=5)


@pytest.mark.timeout(90)
def test_controller_process_labels_nodes(api, tmp_path):
    srv, kubeconfig = api
    # publish slices as the plugin would
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.hal.model import AllocatableDevice
    from k8s_dra_driver_amd.kube.resourceslice import ResourceSlicePublisher

    lib = FakeDeviceLib()
    lib.open()
    ResourceSlicePublisher(
        srv.store, driver_name=DRIVER_NAME, node_name="e2e-node"
    ).publish([AllocatableDevice.from_gpu(g).to_device() for g in lib.enumerate()])

    env = dict(os.environ, PYTHONPATH=REPO, KUBECONFIG=kubeconfig)
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "k8s_dra_driver_amd.controller.main",
            "--poll-interval",
            "0.5",
            "--metrics-port",
            "0",
            "--logging-format",
            "json",
        ],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        _wait_for(
            lambda: "gpu.amd.com/gpu.count"
            in (srv.store.get_node("e2e-node")["metadata"].get("labels") or {}),
            what="controller node labels",
        )
        labels = srv.store.get_node("e2e-node")["metadata"]["labels"]
        assert labels["gpu.amd.com/gpu.count"] == "8"
        assert labels["gpu.amd.com/gpu.architecture"] == "gfx950"
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=5)


def _make_pki(d):
    """CA + server cert (127.0.0.1 SAN) + client cert via openssl."""
    def run(*args):
        subprocess.run(args, cwd=d, check=True, capture_output=True)

    run("openssl", "req", "-x509", "-newkey", "ec", "-pkeyopt",
        "ec_paramgen_curve:P-256", "-keyout", "ca.key", "-out", "ca.crt",
        "-days", "2", "-nodes", "-subj", "/CN=e2e-ca")
    run("openssl", "req", "-newkey", "ec", "-pkeyopt",
        "ec_paramgen_curve:P-256", "-keyout", "server.key", "-out",
        "server.csr", "-nodes", "-subj", "/CN=kubernetes")
    (d / "san.cnf").write_text("subjectAltName=IP:127.0.0.1\n")
    run("openssl", "x509", "-req", "-in", "server.csr", "-CA", "ca.crt",
        "-CAkey", "ca.key", "-CAcreateserial", "-out", "server.crt",
        "-days", "2", "-extfile", "san.cnf")
    run("openssl", "req", "-newkey", "ec", "-pkeyopt",
        "ec_paramgen_curve:P-256", "-keyout", "client.key", "-out",
        "client.csr", "-nodes", "-subj", "/CN=system:node:e2e/O=system:nodes")
    run("openssl", "x509", "-req", "-in", "client.csr", "-CA", "ca.crt",
        "-CAkey", "ca.key", "-CAcreateserial", "-out", "client.crt",
        "-days", "2")


@pytest.mark.timeout(120)
def test_plugin_process_mtls_apiserver(tmp_path):
    """The round-1 gap closed end to end: the REAL plugin
    binary connects to an apiserver that REQUIRES a client certificate,
    using a kind-style kubeconfig with inline cert data, then registers,
    publishes slices and prepares a claim over gRPC."""
    from k8s_dra_driver_amd.kube.miniapiserver import MiniApiServer

    pki = tmp_path / "pki"
    pki.mkdir()
    _make_pki(pki)
    srv = MiniApiServer()
    srv.enable_tls(
        str(pki / "server.crt"),
        str(pki / "server.key"),
        client_ca=str(pki / "ca.crt"),
    )
    srv.start()
    srv.store.put_node({"metadata": {"name": "e2e-node"}})
    kubeconfig = srv.write_kubeconfig(
        str(tmp_path / "kubeconfig"),
        client_cert=str(pki / "client.crt"),
        client_key=str(pki / "client.key"),
    )
    plugin_dir = tmp_path / "plugins" / DRIVER_NAME
    registry_dir = tmp_path / "plugins_registry"
    env = dict(
        os.environ, PYTHONPATH=REPO, NODE_NAME="e2e-node", KUBECONFIG=kubeconfig
    )
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "k8s_dra_driver_amd.plugin.main",
            "--hal",
            "fake",
            "--cdi-root",
            str(tmp_path / "cdi"),
            "--plugin-path",
            str(plugin_dir),
            "--plugin-registration-path",
            str(registry_dir),
        ],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        _wait_for(
            lambda: srv.store.list_resource_slices(DRIVER_NAME),
            what="slice publication over mTLS",
        )
        slices = srv.store.list_resource_slices(DRIVER_NAME)
        assert len(slices[0]["spec"]["devices"]) == 8

        # drive one claim through gRPC to prove the full path
        srv.store.put_resource_claim(
            {
                "metadata": {"namespace": "d", "name": "cm", "uid": "mtls-uid"},
                "status": {
                    "allocation": {
                        "devices": {
                            "results": [
                                {
                                    "request": "gpu",
                                    "driver": DRIVER_NAME,
                                    "pool": "e2e-node",
                                    "device": "gpu-1",
                                }
                            ]
                        }
                    }
                },
            }
        )
        m = V1BETA1
        reg_sock = registry_dir / f"{DRIVER_NAME}.sock"
        reg_channel = grpc.insecure_channel(f"unix://{reg_sock}")
        info = reg_channel.unary_unary(
            f"/{REGISTRATION.service_name}/GetInfo",
            request_serializer=lambda x: x.SerializeToString(),
            response_deserializer=REGISTRATION.PluginInfo.FromString,
        )(REGISTRATION.InfoRequest(), timeout=10)
        reg_channel.close()
        channel = grpc.insecure_channel(f"unix://{info.endpoint}")
        prepare = channel.unary_unary(
            f"/{m.service_name}/NodePrepareResources",
            request_serializer=lambda x: x.SerializeToString(),
            response_deserializer=m.NodePrepareResourcesResponse.FromString,
        )
        req = m.NodePrepareResourcesRequest()
        c = req.claims.add()
        c.namespace, c.name, c.uid = "d", "cm", "mtls-uid"
        resp = prepare(req, timeout=15)
        assert resp.claims["mtls-uid"].error == ""
        channel.close()
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=5)
        srv.stop()


@pytest.mark.timeout(90)
def test_plugin_process_publishes_v1beta2(tmp_path):
    """Real plugin binary against an apiserver preferring v1beta2: the
    published slices carry the flattened Device shape + counters when
    prospective partitioning is on."""
    from k8s_dra_driver_amd.kube.miniapiserver import MiniApiServer

    srv = MiniApiServer()
    srv.store.api_versions = ["v1beta2", "v1beta1"]
    srv.start()
    srv.store.put_node({"metadata": {"name": "e2e-node"}})
    kubeconfig = srv.write_kubeconfig(str(tmp_path / "kubeconfig"))
    env = dict(
        os.environ, PYTHONPATH=REPO, NODE_NAME="e2e-node", KUBECONFIG=kubeconfig
    )
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "k8s_dra_driver_amd.plugin.main",
            "--hal", "fake",
            "--prospective-partitions", "cpx",
            "--cdi-root", str(tmp_path / "cdi"),
            "--plugin-path", str(tmp_path / "plugins" / DRIVER_NAME),
            "--plugin-registration-path", str(tmp_path / "plugins_registry"),
        ],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        _wait_for(
            lambda: srv.store.list_resource_slices(DRIVER_NAME),
            what="v1beta2 slice publication",
        )
        slices = srv.store.list_resource_slices(DRIVER_NAME)
        assert slices[0]["apiVersion"] == "resource.k8s.io/v1beta2"
        devs = [d for s in slices for d in s["spec"]["devices"]]
        assert len(devs) == 8 + 64  # whole GPUs + prospective partitions
        assert all("basic" not in d for d in devs)
        assert any("sharedCounters" in s["spec"] for s in slices)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=5)
        srv.stop()
