"""HTTP KubeClient: talks to a real apiserver over REST (httpx).

The client-go analog of ``pkg/flags/kubeclient.go:30-107``: in-cluster
service-account config or a kubeconfig file (full auth surface — inline
cert data, client TLS, tokenFile, exec plugins — via :mod:`.auth`), with
client-side QPS/burst rate limiting (reference flags kube-api-qps /
kube-api-burst).

Watches carry resourceVersion bookkeeping: each stream starts from a LIST
(whose items are replayed to the handler as synthetic events, so consumers
are level-triggered), then watches from that resourceVersion; reconnects
resume from the last seen version, and a 410 Gone falls back to a fresh
re-list — the client-go reflector contract
(vendor ``resourceslicecontroller.go:407-431`` relies on it).
"""

from __future__ import annotations

import json
import logging
import threading
import time
from typing import Callable, Dict, List, Optional

import httpx

from .auth import KubeConnection, load_in_cluster, load_kubeconfig
from .client import Conflict, KubeClient, NotFound

log = logging.getLogger(__name__)

CORE_V1 = "/api/v1"


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
        context: Optional[str] = None,
        qps: float = 50.0,
        burst: int = 100,
        timeout: float = 30.0,
        connection: Optional[KubeConnection] = None,
    ):
        self._limiter = _RateLimiter(qps, burst)
        self._timeout = timeout
        if connection is not None:
            self._conn = connection
        elif kubeconfig:
            self._conn = load_kubeconfig(kubeconfig, context=context)
        else:
            self._conn = load_in_cluster()
        self._client_lock = threading.Lock()
        self._client_epoch = -1
        self._client: Optional[httpx.Client] = None
        self._resource_base: Optional[str] = None
        self._http()  # fail fast on bad TLS material

    def _http(self) -> httpx.Client:
        """Current HTTP client; rebuilt when the TLS epoch moves (exec
        credential plugins can rotate client certificates)."""
        with self._client_lock:
            if self._client is None or self._client_epoch != self._conn.epoch:
                old = self._client
                self._client = httpx.Client(
                    base_url=self._conn.server,
                    verify=self._conn.ssl_verify(),
                    headers={"Content-Type": "application/json"},
                    timeout=self._timeout,
                )
                self._client_epoch = self._conn.epoch
                if old is not None:
                    old.close()
            return self._client

    def close(self) -> None:
        with self._client_lock:
            if self._client is not None:
                self._client.close()
                self._client = None

    # -- request core ------------------------------------------------------
    def _req(
        self,
        method: str,
        path: str,
        body: Optional[dict] = None,
        headers: Optional[Dict[str, str]] = None,
    ) -> dict:
        self._limiter.acquire()
        h = self._conn.headers()  # may trigger exec refresh -> epoch bump
        if headers:
            h.update(headers)
        r = self._http().request(
            method,
            path,
            content=json.dumps(body) if body is not None else None,
            headers=h,
        )
        if r.status_code == 404:
            raise NotFound(path)
        if r.status_code == 409:
            raise Conflict(path)
        r.raise_for_status()
        return r.json() if r.content else {}

    # -- version negotiation ------------------------------------------------
    def resource_api_versions(self) -> List[str]:
        """Served resource.k8s.io versions, preferred first (apiserver
        group discovery; falls back to v1beta1 when discovery fails)."""
        try:
            out = self._req("GET", "/apis/resource.k8s.io")
            versions = [
                v.get("version")
                for v in out.get("versions", [])
                if v.get("version")
            ]
            pref = (out.get("preferredVersion") or {}).get("version")
            if pref and pref in versions:
                versions.remove(pref)
                versions.insert(0, pref)
            return versions or ["v1beta1"]
        except Exception:
            return ["v1beta1"]

    @property
    def _rbase(self) -> str:
        """REST base for the resource group, negotiated once (newest
        mutually-supported of v1/v1beta2/v1beta1)."""
        if self._resource_base is None:
            served = self.resource_api_versions()
            pick = next(
                (v for v in ("v1", "v1beta2", "v1beta1") if v in served),
                "v1beta1",
            )
            self._resource_base = f"/apis/resource.k8s.io/{pick}"
        return self._resource_base

    # -- KubeClient --------------------------------------------------------
    def get_resource_claim(self, namespace: str, name: str) -> dict:
        return self._req(
            "GET", f"{self._rbase}/namespaces/{namespace}/resourceclaims/{name}"
        )

    def create_resource_slice(self, obj: dict) -> dict:
        return self._req("POST", f"{self._rbase}/resourceslices", obj)

    def update_resource_slice(self, obj: dict) -> dict:
        name = obj["metadata"]["name"]
        return self._req("PUT", f"{self._rbase}/resourceslices/{name}", obj)

    def delete_resource_slice(self, name: str) -> None:
        try:
            self._req("DELETE", f"{self._rbase}/resourceslices/{name}")
        except NotFound:
            pass

    def list_resource_slices(self, driver: Optional[str] = None) -> List[dict]:
        params = ""
        if driver:
            params = f"?fieldSelector=spec.driver%3D{driver}"
        out = self._req("GET", f"{self._rbase}/resourceslices{params}")
        return out.get("items", [])

    def list_resource_claims(self) -> List[dict]:
        out = self._req("GET", f"{self._rbase}/resourceclaims")
        return out.get("items", [])

    def update_resource_claim_status(self, obj: dict) -> dict:
        ns = obj["metadata"].get("namespace", "default")
        name = obj["metadata"]["name"]
        return self._req(
            "PUT",
            f"{self._rbase}/namespaces/{ns}/resourceclaims/{name}/status",
            obj,
        )

    def get_device_classes(self) -> List[dict]:
        out = self._req("GET", f"{self._rbase}/deviceclasses")
        return out.get("items", [])

    # -- watches -----------------------------------------------------------
    def _watch(
        self,
        collection_path: str,
        handler: Callable[[str, dict], None],
        *,
        name: str,
    ):
        """List+watch with resourceVersion resume (client-go reflector
        shape). The initial LIST (and every re-list after a dropped window)
        is replayed to the handler as synthetic ADDED events, so consumers
        see a level-triggered stream and self-heal across gaps."""
        stop_event = threading.Event()
        outer = self

        def list_and_rv() -> Optional[str]:
            try:
                out = outer._req("GET", collection_path)
            except Exception as e:
                log.debug("%s: re-list failed: %s", name, e)
                return None
            for item in out.get("items", []):
                handler("ADDED", item)
            return (out.get("metadata") or {}).get("resourceVersion") or ""

        def run():
            rv: Optional[str] = None
            while not stop_event.is_set():
                if rv is None:
                    rv = list_and_rv()
                    if rv is None:
                        if stop_event.wait(1.0):
                            return
                        continue
                params = "?watch=true&allowWatchBookmarks=true"
                if rv:
                    params += f"&resourceVersion={rv}"
                try:
                    h = outer._conn.headers()
                    with outer._http().stream(
                        "GET",
                        f"{collection_path}{params}",
                        headers=h,
                        timeout=httpx.Timeout(5.0, read=None),
                    ) as r:
                        if r.status_code == 410:
                            rv = None  # window expired: re-list
                            continue
                        r.raise_for_status()
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
                            etype = ev.get("type", "")
                            obj = ev.get("object") or {}
                            new_rv = (obj.get("metadata") or {}).get(
                                "resourceVersion"
                            )
                            if etype == "ERROR":
                                # typically 410 Gone mid-stream
                                rv = None
                                break
                            if new_rv:
                                rv = new_rv
                            if etype == "BOOKMARK":
                                continue
                            handler(etype, obj)
                except Exception:
                    if stop_event.is_set():
                        return
                    time.sleep(1.0)  # transient-error retry (imex.go parity)

        thread = threading.Thread(target=run, name=name, daemon=True)
        thread.start()

        class _Watch:
            def stop(self_inner) -> None:
                stop_event.set()

        return _Watch()

    def watch_resource_claims(self, handler):
        return self._watch(
            f"{self._rbase}/resourceclaims", handler, name="claims-watch"
        )

    def watch_resource_slices(self, handler):
        return self._watch(
            f"{self._rbase}/resourceslices", handler, name="slices-watch"
        )

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
        h = self._conn.headers()
        h["Content-Type"] = "application/strategic-merge-patch+json"
        r = self._http().request(
            "PATCH",
            f"{CORE_V1}/nodes/{name}",
            content=json.dumps(body),
            headers=h,
        )
        if r.status_code == 404:
            raise NotFound(name)
        r.raise_for_status()
        return r.json()
