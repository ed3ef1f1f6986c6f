"""Minimal Kubernetes API client interface + in-memory fake.

The reference uses client-go ClientSets built by ``pkg/flags/kubeclient.go``.
Here the driver needs a narrow surface: ResourceClaim reads (the prepare hot
path does one GET per claim, reference ``driver.go:122``), ResourceSlice
CRUD (publication), and node label reads/writes (controller). The
``KubeClient`` ABC captures exactly that; ``InMemoryKube`` is the
zero-cluster fake used by tests and the bench harness (the apiserver analog
of the fake HAL); ``HttpKube`` talks to a real apiserver via the REST API
(httpx) for in-cluster deployment.
"""

from __future__ import annotations

import abc
import copy
import threading
from typing import Callable, Dict, List, Optional


class NotFound(KeyError):
    pass


class Conflict(RuntimeError):
    pass


class KubeClient(abc.ABC):
    @abc.abstractmethod
    def get_resource_claim(self, namespace: str, name: str) -> dict: ...

    def list_resource_claims(self) -> List[dict]:
        """All ResourceClaims (controller allocator input); optional."""
        raise NotImplementedError

    def update_resource_claim_status(self, obj: dict) -> dict:
        """Write claim.status (allocation); optional."""
        raise NotImplementedError

    def get_device_classes(self) -> List[dict]:
        """DeviceClass objects (controller allocator input); optional."""
        return []

    def resource_api_versions(self) -> List[str]:
        """Served versions of the resource.k8s.io group, preferred first
        (apiserver discovery). Drives version-negotiated ResourceSlice
        publication (reference draplugin.go:342-350 multi-version
        pattern; K8s 1.33 flattens the Device shape in v1beta2)."""
        return ["v1beta1"]

    def create_event(self, namespace: str, event: dict) -> None:
        """Emit a Kubernetes Event (operator visibility); best-effort."""

    def watch_resource_claims(self, handler: Callable[[str, dict], None]):
        """Subscribe to ResourceClaim changes: ``handler(event_type, obj)``
        with event_type ADDED|MODIFIED|DELETED. Returns an object with
        ``stop()``, or None if watching is unsupported (callers fall back
        to polling) — the client-go informer analog (reference
        imex.go:222-295)."""
        return None

    def watch_resource_slices(self, handler: Callable[[str, dict], None]):
        """Subscribe to ResourceSlice changes (drift self-heal input,
        reference resourceslicecontroller.go:407-431). Same contract as
        :meth:`watch_resource_claims`; None if unsupported."""
        return None

    @abc.abstractmethod
    def create_resource_slice(self, obj: dict) -> dict: ...

    @abc.abstractmethod
    def update_resource_slice(self, obj: dict) -> dict: ...

    @abc.abstractmethod
    def delete_resource_slice(self, name: str) -> None: ...

    @abc.abstractmethod
    def list_resource_slices(self, driver: Optional[str] = None) -> List[dict]: ...

    @abc.abstractmethod
    def get_node(self, name: str) -> dict: ...

    @abc.abstractmethod
    def patch_node_labels(self, name: str, labels: Dict[str, Optional[str]]) -> dict: ...


class InMemoryKube(KubeClient):
    """Thread-safe in-memory apiserver good enough for plugin + controller
    tests and the latency bench (no network, no serialization overhead —
    the measured path is the driver, not a fake wire)."""

    def __init__(self):
        self._lock = threading.RLock()
        #: served resource.k8s.io versions, preferred first (tests flip
        #: this to ["v1beta2", "v1beta1"] to exercise negotiation)
        self.api_versions: List[str] = ["v1beta1"]
        self.resource_claims: Dict[str, dict] = {}  # key: ns/name
        self.resource_slices: Dict[str, dict] = {}
        self.nodes: Dict[str, dict] = {}
        self.device_classes: Dict[str, dict] = {}
        self.events: List[dict] = []
        self._rv = 0
        #: watch hooks: fn(kind, verb, obj)
        self.watchers: List[Callable[[str, str, dict], None]] = []

    # -- helpers -----------------------------------------------------------
    def _next_rv(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _emit(self, kind: str, verb: str, obj: dict) -> None:
        for fn in list(self.watchers):
            fn(kind, verb, obj)

    def put_resource_claim(self, obj: dict) -> dict:
        with self._lock:
            key = f"{obj['metadata'].get('namespace','default')}/{obj['metadata']['name']}"
            verb = "MODIFIED" if key in self.resource_claims else "ADDED"
            obj["metadata"]["resourceVersion"] = self._next_rv()
            self.resource_claims[key] = copy.deepcopy(obj)
        self._emit("ResourceClaim", verb, obj)
        return obj

    def put_node(self, obj: dict) -> dict:
        with self._lock:
            obj.setdefault("metadata", {}).setdefault("labels", {})
            self.nodes[obj["metadata"]["name"]] = copy.deepcopy(obj)
            self._emit("Node", "put", obj)
            return obj

    # -- KubeClient --------------------------------------------------------
    def get_resource_claim(self, namespace: str, name: str) -> dict:
        with self._lock:
            key = f"{namespace}/{name}"
            if key not in self.resource_claims:
                raise NotFound(f"resourceclaim {key}")
            return copy.deepcopy(self.resource_claims[key])

    def list_resource_claims(self) -> List[dict]:
        with self._lock:
            return [copy.deepcopy(c) for c in self.resource_claims.values()]

    def update_resource_claim_status(self, obj: dict) -> dict:
        with self._lock:
            key = f"{obj['metadata'].get('namespace','default')}/{obj['metadata']['name']}"
            if key not in self.resource_claims:
                raise NotFound(f"resourceclaim {key}")
            cur = self.resource_claims[key]
            cur["status"] = copy.deepcopy(obj.get("status") or {})
            cur["metadata"]["resourceVersion"] = self._next_rv()
        self._emit("ResourceClaim", "status", cur)
        return copy.deepcopy(cur)

    def get_device_classes(self) -> List[dict]:
        with self._lock:
            return [copy.deepcopy(d) for d in self.device_classes.values()]

    def put_device_class(self, obj: dict) -> dict:
        with self._lock:
            self.device_classes[obj["metadata"]["name"]] = copy.deepcopy(obj)
            return obj

    def resource_api_versions(self) -> List[str]:
        return list(self.api_versions)

    def create_event(self, namespace: str, event: dict) -> None:
        with self._lock:
            self.events.append({"namespace": namespace, **copy.deepcopy(event)})

    def watch_resource_claims(self, handler: Callable[[str, dict], None]):
        def hook(kind: str, verb: str, obj: dict) -> None:
            if kind == "ResourceClaim" and verb in ("ADDED", "MODIFIED", "DELETED"):
                handler(verb, copy.deepcopy(obj))

        self.watchers.append(hook)
        kube = self

        class _Watch:
            def stop(self_inner) -> None:
                if hook in kube.watchers:
                    kube.watchers.remove(hook)

        return _Watch()

    def watch_resource_slices(self, handler: Callable[[str, dict], None]):
        verb_map = {"create": "ADDED", "update": "MODIFIED", "delete": "DELETED"}

        def hook(kind: str, verb: str, obj: dict) -> None:
            if kind == "ResourceSlice" and verb in verb_map:
                handler(verb_map[verb], copy.deepcopy(obj))

        self.watchers.append(hook)
        kube = self

        class _Watch:
            def stop(self_inner) -> None:
                if hook in kube.watchers:
                    kube.watchers.remove(hook)

        return _Watch()

    def create_resource_slice(self, obj: dict) -> dict:
        # events are emitted OUTSIDE the store lock: a synchronous watcher
        # (e.g. the publisher's drift self-heal) re-enters the store from
        # its own lock, and emitting under _lock would be an ABBA deadlock
        with self._lock:
            name = obj["metadata"]["name"]
            if name in self.resource_slices:
                raise Conflict(f"resourceslice {name} exists")
            obj["metadata"]["resourceVersion"] = self._next_rv()
            self.resource_slices[name] = copy.deepcopy(obj)
        self._emit("ResourceSlice", "create", obj)
        return copy.deepcopy(obj)

    def update_resource_slice(self, obj: dict) -> dict:
        with self._lock:
            name = obj["metadata"]["name"]
            if name not in self.resource_slices:
                raise NotFound(f"resourceslice {name}")
            cur = self.resource_slices[name]
            rv = obj["metadata"].get("resourceVersion")
            if rv and rv != cur["metadata"]["resourceVersion"]:
                raise Conflict(f"resourceslice {name} resourceVersion mismatch")
            obj["metadata"]["resourceVersion"] = self._next_rv()
            self.resource_slices[name] = copy.deepcopy(obj)
        self._emit("ResourceSlice", "update", obj)
        return copy.deepcopy(obj)

    def delete_resource_slice(self, name: str) -> None:
        with self._lock:
            obj = self.resource_slices.pop(name, None)
        if obj is not None:
            self._emit("ResourceSlice", "delete", obj)

    def list_resource_slices(self, driver: Optional[str] = None) -> List[dict]:
        with self._lock:
            out = [copy.deepcopy(s) for s in self.resource_slices.values()]
        if driver:
            out = [s for s in out if s.get("spec", {}).get("driver") == driver]
        return out

    def get_node(self, name: str) -> dict:
        with self._lock:
            if name not in self.nodes:
                raise NotFound(f"node {name}")
            return copy.deepcopy(self.nodes[name])

    def patch_node_labels(self, name: str, labels: Dict[str, Optional[str]]) -> dict:
        with self._lock:
            node = self.nodes.get(name)
            if node is None:
                raise NotFound(f"node {name}")
            lbl = node["metadata"].setdefault("labels", {})
            for k, v in labels.items():
                if v is None:
                    lbl.pop(k, None)
                else:
                    lbl[k] = v
            self._emit("Node", "patch", node)
            return copy.deepcopy(node)
