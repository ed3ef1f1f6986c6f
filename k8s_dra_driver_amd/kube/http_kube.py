"""HTTP KubeClient: talks to a real apiserver over REST (httpx).

The client-go analog of ``pkg/flags/kubeclient.go:30-107``: in-cluster
service-account config or a kubeconfig file, with client-side QPS/burst
rate limiting (reference flags kube-api-qps/kube-api-burst).
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional

import httpx
import yaml

from .client import Conflict, KubeClient, NotFound

RESOURCE_V1BETA1 = "/apis/resource.k8s.io/v1beta1"
CORE_V1 = "/api/v1"

SA_TOKEN = "/var/run/secrets/kubernetes.io/serviceaccount/token"
SA_CA = "/var/run/secrets/kubernetes.io/serviceaccount/ca.crt"


class _RateLimiter:
    """Token bucket: qps refill, burst capacity."""

    def __init__(self, qps: float, burst: int):
        self.qps = max(qps, 0.001)
        self.burst = max(burst, 1)
        self._tokens = float(burst)
        self._last = time.monotonic()
        self._lock = threading.Lock()

    def acquire(self) -> None:
        while True:
            with self._lock:
                now = time.monotonic()
                self._tokens = min(
                    self.burst, self._tokens + (now - self._last) * self.qps
                )
                self._last = now
                if self._tokens >= 1:
                    self._tokens -= 1
                    return
                wait = (1 - self._tokens) / self.qps
            time.sleep(wait)


class HttpKube(KubeClient):
    def __init__(
        self,
        kubeconfig: Optional[str] = None,
        *,
        qps: float = 50.0,
        burst: int = 100,
        timeout: float = 30.0,
    ):
        self._limiter = _RateLimiter(qps, burst)
        headers = {"Content-Type": "application/json"}
        if kubeconfig:
            base_url, verify, headers2 = self._from_kubeconfig(kubeconfig)
            headers.update(headers2)
        else:
            base_url, verify, headers2 = self._in_cluster()
            headers.update(headers2)
        self._client = httpx.Client(
            base_url=base_url, verify=verify, headers=headers, timeout=timeout
        )

    @staticmethod
    def _in_cluster():
        host = os.environ.get("KUBERNETES_SERVICE_HOST")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        if not host:
            raise RuntimeError(
                "not running in-cluster (KUBERNETES_SERVICE_HOST unset) and "
                "no kubeconfig given"
            )
        with open(SA_TOKEN) as f:
            token = f.read().strip()
        return (
            f"https://{host}:{port}",
            SA_CA if os.path.exists(SA_CA) else False,
            {"Authorization": f"Bearer {token}"},
        )

    @staticmethod
    def _from_kubeconfig(path: str):
        with open(path) as f:
            cfg = yaml.safe_load(f)
        ctx_name = cfg.get("current-context")
        ctx = next(c for c in cfg["contexts"] if c["name"] == ctx_name)["context"]
        cluster = next(
            c for c in cfg["clusters"] if c["name"] == ctx["cluster"]
        )["cluster"]
        user = next(u for u in cfg["users"] if u["name"] == ctx["user"])["user"]
        headers: Dict[str, str] = {}
        if "token" in user:
            headers["Authorization"] = f"Bearer {user['token']}"
        verify = cluster.get("certificate-authority", True)
        if cluster.get("insecure-skip-tls-verify"):
            verify = False
        return cluster["server"], verify, headers

    # -- request core ------------------------------------------------------
    def _req(self, method: str, path: str, body: Optional[dict] = None) -> dict:
        self._limiter.acquire()
        r = self._client.request(
            method, path, content=json.dumps(body) if body is not None else None
        )
        if r.status_code == 404:
            raise NotFound(path)
        if r.status_code == 409:
            raise Conflict(path)
        r.raise_for_status()
        return r.json() if r.content else {}

    # -- KubeClient --------------------------------------------------------
    def get_resource_claim(self, namespace: str, name: str) -> dict:
        return self._req(
            "GET", f"{RESOURCE_V1BETA1}/namespaces/{namespace}/resourceclaims/{name}"
        )

    def create_resource_slice(self, obj: dict) -> dict:
        return self._req("POST", f"{RESOURCE_V1BETA1}/resourceslices", obj)

    def update_resource_slice(self, obj: dict) -> dict:
        name = obj["metadata"]["name"]
        return self._req("PUT", f"{RESOURCE_V1BETA1}/resourceslices/{name}", obj)

    def delete_resource_slice(self, name: str) -> None:
        try:
            self._req("DELETE", f"{RESOURCE_V1BETA1}/resourceslices/{name}")
        except NotFound:
            pass

    def list_resource_slices(self, driver: Optional[str] = None) -> List[dict]:
        params = ""
        if driver:
            params = f"?fieldSelector=spec.driver%3D{driver}"
        out = self._req("GET", f"{RESOURCE_V1BETA1}/resourceslices{params}")
        return out.get("items", [])

    def list_resource_claims(self) -> List[dict]:
        out = self._req("GET", f"{RESOURCE_V1BETA1}/resourceclaims")
        return out.get("items", [])

    def update_resource_claim_status(self, obj: dict) -> dict:
        ns = obj["metadata"].get("namespace", "default")
        name = obj["metadata"]["name"]
        return self._req(
            "PUT",
            f"{RESOURCE_V1BETA1}/namespaces/{ns}/resourceclaims/{name}/status",
            obj,
        )

    def get_device_classes(self) -> List[dict]:
        out = self._req("GET", f"{RESOURCE_V1BETA1}/deviceclasses")
        return out.get("items", [])

    def watch_resource_claims(self, handler):
        """Streaming watch (?watch=true, JSON-line events) with reconnect —
        the informer analog. stop() is lazy: the reader exits at the next
        event or when the connection drops."""
        stop_event = threading.Event()
        client = self._client

        def run():
            while not stop_event.is_set():
                try:
                    with client.stream(
                        "GET",
                        f"{RESOURCE_V1BETA1}/resourceclaims?watch=true",
                        timeout=httpx.Timeout(5.0, read=None),
                    ) as r:
                        for line in r.iter_lines():
                            if stop_event.is_set():
                                return
                            line = line.strip()
                            if not line:
                                continue
                            try:
                                ev = json.loads(line)
                            except json.JSONDecodeError:
                                continue
                            handler(ev.get("type", ""), ev.get("object") or {})
                except Exception:
                    if stop_event.is_set():
                        return
                    time.sleep(1.0)  # transient-error retry (imex.go parity)

        thread = threading.Thread(target=run, name="claims-watch", daemon=True)
        thread.start()

        class _Watch:
            def stop(self_inner) -> None:
                stop_event.set()

        return _Watch()

    def create_event(self, namespace: str, event: dict) -> None:
        try:
            self._req("POST", f"{CORE_V1}/namespaces/{namespace}/events", event)
        except Exception:
            pass  # events are best-effort

    def get_node(self, name: str) -> dict:
        return self._req("GET", f"{CORE_V1}/nodes/{name}")

    def patch_node_labels(self, name: str, labels: Dict[str, Optional[str]]) -> dict:
        self._limiter.acquire()
        body = {"metadata": {"labels": labels}}
        r = self._client.request(
            "PATCH",
            f"{CORE_V1}/nodes/{name}",
            content=json.dumps(body),
            headers={"Content-Type": "application/strategic-merge-patch+json"},
        )
        if r.status_code == 404:
            raise NotFound(name)
        r.raise_for_status()
        return r.json()
