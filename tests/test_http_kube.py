"""HttpKube tests against a minimal in-process apiserver (HTTP)."""

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from k8s_dra_driver_amd.kube.client import Conflict, NotFound
from k8s_dra_driver_amd.kube.http_kube import HttpKube, _RateLimiter


class MiniApiServer:
    """Just enough apiserver: resourceclaims GET, resourceslices CRUD,
    nodes GET/PATCH."""

    def __init__(self):
        self.claims = {}
        self.slices = {}
        self.nodes = {}
        self.requests = []
        self.auth_headers = []
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _json(self, code, obj):
                data = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

            def _body(self):
                n = int(self.headers.get("Content-Length") or 0)
                return json.loads(self.rfile.read(n)) if n else None

            def do_GET(self):
                outer.requests.append(("GET", self.path))
                outer.auth_headers.append(self.headers.get("Authorization"))
                parts = self.path.split("?")[0].strip("/").split("/")
                if "resourceclaims" in parts:
                    ns, name = parts[4], parts[6]
                    obj = outer.claims.get(f"{ns}/{name}")
                    return self._json(200, obj) if obj else self._json(404, {})
                if parts[-1] == "resourceslices":
                    return self._json(200, {"items": list(outer.slices.values())})
                if "nodes" in parts:
                    obj = outer.nodes.get(parts[-1])
                    return self._json(200, obj) if obj else self._json(404, {})
                self._json(404, {})

            def do_POST(self):
                outer.requests.append(("POST", self.path))
                obj = self._body()
                name = obj["metadata"]["name"]
                if name in outer.slices:
                    return self._json(409, {})
                outer.slices[name] = obj
                self._json(201, obj)

            def do_PUT(self):
                obj = self._body()
                name = obj["metadata"]["name"]
                if name not in outer.slices:
                    return self._json(404, {})
                outer.slices[name] = obj
                self._json(200, obj)

            def do_DELETE(self):
                name = self.path.strip("/").split("/")[-1]
                if outer.slices.pop(name, None) is None:
                    return self._json(404, {})
                self._json(200, {})

            def do_PATCH(self):
                name = self.path.strip("/").split("/")[-1]
                node = outer.nodes.get(name)
                if node is None:
                    return self._json(404, {})
                patch = self._body()
                for k, v in patch["metadata"]["labels"].items():
                    if v is None:
                        node["metadata"].setdefault("labels", {}).pop(k, None)
                    else:
                        node["metadata"].setdefault("labels", {})[k] = v
                self._json(200, node)

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.thread = threading.Thread(
            target=self.server.serve_forever, daemon=True
        )
        self.thread.start()

    @property
    def url(self):
        return f"http://127.0.0.1:{self.server.server_address[1]}"

    def stop(self):
        self.server.shutdown()


@pytest.fixture
def api(tmp_path, monkeypatch):
    srv = MiniApiServer()
    # kubeconfig pointing at the mini server
    kc = tmp_path / "kubeconfig"
    kc.write_text(
        json.dumps(
            {
                "current-context": "c",
                "contexts": [{"name": "c", "context": {"cluster": "k", "user": "u"}}],
                "clusters": [{"name": "k", "cluster": {"server": srv.url}}],
                "users": [{"name": "u", "user": {"token": "test-token"}}],
            }
        )
    )
    client = HttpKube(kubeconfig=str(kc), qps=1000, burst=1000)
    yield srv, client
    srv.stop()


def test_resource_claim_get(api):
    srv, client = api
    srv.claims["default/c1"] = {"metadata": {"name": "c1", "uid": "u1"}}
    got = client.get_resource_claim("default", "c1")
    assert got["metadata"]["uid"] == "u1"
    with pytest.raises(NotFound):
        client.get_resource_claim("default", "nope")


def test_resource_slice_crud(api):
    srv, client = api
    obj = {"metadata": {"name": "s1"}, "spec": {"driver": "gpu.amd.com"}}
    client.create_resource_slice(obj)
    with pytest.raises(Conflict):
        client.create_resource_slice(obj)
    obj["spec"]["x"] = 1
    client.update_resource_slice(obj)
    assert client.list_resource_slices("gpu.amd.com")[0]["spec"]["x"] == 1
    client.delete_resource_slice("s1")
    client.delete_resource_slice("s1")  # tolerated
    assert client.list_resource_slices() == []


def test_node_label_patch(api):
    srv, client = api
    srv.nodes["n1"] = {"metadata": {"name": "n1", "labels": {}}}
    client.patch_node_labels("n1", {"a": "1", "b": None})
    assert srv.nodes["n1"]["metadata"]["labels"] == {"a": "1"}


def test_bearer_token_sent(api):
    srv, client = api
    srv.nodes["n1"] = {"metadata": {"name": "n1"}}
    client.get_node("n1")
    # our kubeconfig carries a token; sent per-request (auth providers can
    # rotate credentials, so headers are no longer baked into the client)
    assert srv.auth_headers[-1] == "Bearer test-token"


def test_rate_limiter_blocks():
    import time

    rl = _RateLimiter(qps=100, burst=2)
    rl.acquire()
    rl.acquire()
    t0 = time.monotonic()
    rl.acquire()  # must wait ~10ms for a token
    assert time.monotonic() - t0 > 0.005
