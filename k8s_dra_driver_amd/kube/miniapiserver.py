"""Mini apiserver: a tiny HTTP server speaking just enough of the
Kubernetes REST API for this driver's binaries to run without a cluster.

Powers the multi-process e2e tests and clusterless demos: start it, point
``amd-dra-kubeletplugin --kubeconfig <generated>`` and
``amd-dra-controller`` at it, and drive claims end-to-end. State is the
in-memory :class:`k8s_dra_driver_amd.kube.client.InMemoryKube` store, so
assertions can inspect it directly in-process.
"""

from __future__ import annotations

import json
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from .client import Conflict, InMemoryKube, NotFound

_CLAIM_RE = re.compile(
    r"^/apis/resource\.k8s\.io/(?:v1beta[12]|v1)/namespaces/([^/]+)/resourceclaims/([^/]+?)(/status)?$"
)
_SLICE_RE = re.compile(r"^/apis/resource\.k8s\.io/(?:v1beta[12]|v1)/resourceslices(?:/([^/]+))?$")
_CLAIMS_RE = re.compile(r"^/apis/resource\.k8s\.io/(?:v1beta[12]|v1)/resourceclaims$")
_CLASSES_RE = re.compile(r"^/apis/resource\.k8s\.io/(?:v1beta[12]|v1)/deviceclasses$")
_GROUP_DISCOVERY = "/apis/resource.k8s.io"
_NODE_RE = re.compile(r"^/api/v1/nodes/([^/]+)$")


class MiniApiServer:
    def __init__(self, store: Optional[InMemoryKube] = None, port: int = 0):
        self.store = store or InMemoryKube()
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

            def _list(self, items):
                return self._json(
                    200,
                    {
                        "metadata": {"resourceVersion": str(outer.store._rv)},
                        "items": items,
                    },
                )

            def do_GET(self):
                path = self.path.split("?")[0]
                if path == _GROUP_DISCOVERY:
                    versions = outer.store.api_versions
                    return self._json(
                        200,
                        {
                            "kind": "APIGroup",
                            "name": "resource.k8s.io",
                            "versions": [
                                {
                                    "groupVersion": f"resource.k8s.io/{v}",
                                    "version": v,
                                }
                                for v in versions
                            ],
                            "preferredVersion": {
                                "groupVersion": f"resource.k8s.io/{versions[0]}",
                                "version": versions[0],
                            },
                        },
                    )
                if _CLAIMS_RE.match(path) and "watch=true" in self.path:
                    return self._watch_kind("ResourceClaim")
                m = _CLAIM_RE.match(path)
                if m:
                    try:
                        return self._json(
                            200, outer.store.get_resource_claim(m.group(1), m.group(2))
                        )
                    except NotFound:
                        return self._json(404, {"reason": "NotFound"})
                if _CLAIMS_RE.match(path):
                    return self._list(outer.store.list_resource_claims())
                if _CLASSES_RE.match(path):
                    return self._list(outer.store.get_device_classes())
                m = _SLICE_RE.match(path)
                if m and not m.group(1):
                    if "watch=true" in self.path:
                        return self._watch_kind("ResourceSlice")
                    return self._list(outer.store.list_resource_slices())
                m = _NODE_RE.match(path)
                if m:
                    try:
                        return self._json(200, outer.store.get_node(m.group(1)))
                    except NotFound:
                        return self._json(404, {})
                self._json(404, {"path": path})

            def _watch_kind(self, want_kind):
                """Streamed watch: JSON-line events until client disconnect
                (read-until-close framing)."""
                import queue

                q: "queue.Queue" = queue.Queue()
                verb_map = {
                    "ADDED": "ADDED",
                    "MODIFIED": "MODIFIED",
                    "DELETED": "DELETED",
                    "status": "MODIFIED",
                    "create": "ADDED",
                    "update": "MODIFIED",
                    "delete": "DELETED",
                }

                def hook(kind, verb, obj):
                    if kind == want_kind and verb in verb_map:
                        q.put({"type": verb_map[verb], "object": obj})

                outer.store.watchers.append(hook)
                try:
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("Connection", "close")
                    self.end_headers()
                    while True:
                        try:
                            ev = q.get(timeout=0.5)
                        except queue.Empty:
                            continue
                        self.wfile.write((json.dumps(ev) + "\n").encode())
                        self.wfile.flush()
                except (BrokenPipeError, ConnectionResetError):
                    pass
                finally:
                    if hook in outer.store.watchers:
                        outer.store.watchers.remove(hook)

            def do_POST(self):
                path = self.path.split("?")[0]
                if _SLICE_RE.match(path):
                    try:
                        return self._json(
                            201, outer.store.create_resource_slice(self._body())
                        )
                    except Conflict:
                        return self._json(409, {})
                self._json(404, {})

            def do_PUT(self):
                path = self.path.split("?")[0]
                m = _CLAIM_RE.match(path)
                if m and m.group(3):  # /status
                    try:
                        return self._json(
                            200,
                            outer.store.update_resource_claim_status(self._body()),
                        )
                    except NotFound:
                        return self._json(404, {})
                m = _SLICE_RE.match(path)
                if m and m.group(1):
                    try:
                        return self._json(
                            200, outer.store.update_resource_slice(self._body())
                        )
                    except NotFound:
                        return self._json(404, {})
                    except Conflict:
                        return self._json(409, {})
                self._json(404, {})

            def do_DELETE(self):
                path = self.path.split("?")[0]
                m = _SLICE_RE.match(path)
                if m and m.group(1):
                    outer.store.delete_resource_slice(m.group(1))
                    return self._json(200, {})
                self._json(404, {})

            def do_PATCH(self):
                path = self.path.split("?")[0]
                m = _NODE_RE.match(path)
                if m:
                    patch = self._body()
                    try:
                        return self._json(
                            200,
                            outer.store.patch_node_labels(
                                m.group(1),
                                (patch.get("metadata") or {}).get("labels") or {},
                            ),
                        )
                    except NotFound:
                        return self._json(404, {})
                self._json(404, {})

        self._server = ThreadingHTTPServer(("127.0.0.1", port), Handler)
        self._thread: Optional[threading.Thread] = None
        self._tls: Optional[dict] = None

    def enable_tls(
        self,
        server_cert: str,
        server_key: str,
        *,
        client_ca: Optional[str] = None,
        ca_file: Optional[str] = None,
    ) -> "MiniApiServer":
        """Serve HTTPS like a kubeadm/kind apiserver; with ``client_ca``
        a client certificate is REQUIRED (mTLS), which is exactly the
        connection shape round-1's HttpKube could not make."""
        import ssl

        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(server_cert, server_key)
        if client_ca:
            ctx.load_verify_locations(client_ca)
            ctx.verify_mode = ssl.CERT_REQUIRED
        self._server.socket = ctx.wrap_socket(
            self._server.socket, server_side=True
        )
        self._tls = {"ca_file": ca_file or client_ca}
        return self

    @property
    def url(self) -> str:
        scheme = "https" if self._tls else "http"
        return f"{scheme}://127.0.0.1:{self._server.server_address[1]}"

    def write_kubeconfig(
        self,
        path: str,
        *,
        client_cert: Optional[str] = None,
        client_key: Optional[str] = None,
    ) -> str:
        """Token kubeconfig by default; kind-style inline client-cert +
        CA data when TLS is enabled and cert paths are given."""
        import base64

        cluster: dict = {"server": self.url}
        user: dict = {"token": "dev"}
        if self._tls and self._tls.get("ca_file"):
            with open(self._tls["ca_file"], "rb") as f:
                cluster["certificate-authority-data"] = base64.b64encode(
                    f.read()
                ).decode()
        if client_cert and client_key:
            with open(client_cert, "rb") as f:
                cert_b64 = base64.b64encode(f.read()).decode()
            with open(client_key, "rb") as f:
                key_b64 = base64.b64encode(f.read()).decode()
            user = {
                "client-certificate-data": cert_b64,
                "client-key-data": key_b64,
            }
        with open(path, "w") as f:
            json.dump(
                {
                    "current-context": "mini",
                    "contexts": [
                        {"name": "mini", "context": {"cluster": "mini", "user": "u"}}
                    ],
                    "clusters": [{"name": "mini", "cluster": cluster}],
                    "users": [{"name": "u", "user": user}],
                },
                f,
            )
        return path

    def start(self) -> "MiniApiServer":
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="mini-apiserver", daemon=True
        )
        self._thread.start()
        return self

    def stop(self) -> None:
        self._server.shutdown()
        if self._thread:
            self._thread.join(timeout=5)
