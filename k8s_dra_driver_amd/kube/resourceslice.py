"""ResourceSlice publication: reconcile allocatable devices to the apiserver.

The reference delegates to the vendored workqueue-based
``resourceslice.Controller`` (``vendor/k8s.io/dynamic-resource-allocation/
resourceslice/resourceslicecontroller.go:74-157``). This is the same
reconcile contract in-process:

- devices are published in pools of <= MAX_DEVICES_PER_SLICE (the reference
  publishes 128 IMEX channels per slice, ``imex.go:43``);
- the pool generation bumps whenever the device set changes (repartition!)
  so the scheduler discards stale slices;
- reconcile is level-triggered and idempotent: publish() computes desired
  slices and creates/updates/deletes to match.
"""

from __future__ import annotations

import hashlib
import json
import threading
from typing import List, Optional

from .client import Conflict, KubeClient, NotFound

MAX_DEVICES_PER_SLICE = 128
API_VERSION = "resource.k8s.io/v1beta1"


def _devices_fingerprint(devices: List[dict]) -> str:
    data = json.dumps(devices, sort_keys=True).encode()
    return hashlib.sha256(data).hexdigest()[:16]


class ResourceSlicePublisher:
    def __init__(
        self,
        client: KubeClient,
        *,
        driver_name: str,
        node_name: str,
        pool_name: Optional[str] = None,
        node_uid: str = "",
    ):
        self.client = client
        self.driver_name = driver_name
        self.node_name = node_name
        self.pool_name = pool_name or node_name
        # With a node UID, slices carry an ownerReference so apiserver GC
        # deletes them when the Node object goes away (the k8s-native
        # complement to the controller's orphan sweep).
        self.node_uid = node_uid
        self._lock = threading.Lock()
        self._generation = 0
        self._last_fingerprint: Optional[str] = None

    def _slice_name(self, index: int) -> str:
        safe_driver = self.driver_name.replace("/", "-").replace(".", "-")
        return f"{self.node_name}-{safe_driver}-{index}"

    def publish(self, devices: List[dict]) -> List[dict]:
        """Reconcile the published slices to carry exactly ``devices``.

        Returns the slice objects as stored. No-op (no API calls) when the
        device set is unchanged — cheap to call after every enumeration.
        """
        with self._lock:
            fp = _devices_fingerprint(devices)
            existing = {
                s["metadata"]["name"]: s
                for s in self.client.list_resource_slices(self.driver_name)
                if s["spec"].get("nodeName") == self.node_name
            }
            if fp == self._last_fingerprint and existing:
                return list(existing.values())
            if self._generation == 0 and existing:
                # plugin restart: never regress the pool generation below
                # what is already published (the scheduler treats higher
                # generations as authoritative for multi-slice pools)
                self._generation = max(
                    (s["spec"].get("pool", {}).get("generation", 0))
                    for s in existing.values()
                )
            self._generation += 1

            chunks = [
                devices[i : i + MAX_DEVICES_PER_SLICE]
                for i in range(0, len(devices), MAX_DEVICES_PER_SLICE)
            ] or [[]]
            desired_names = {self._slice_name(i) for i in range(len(chunks))}
            out = []
            for i, chunk in enumerate(chunks):
                name = self._slice_name(i)
                meta: dict = {"name": name}
                if self.node_uid:
                    meta["ownerReferences"] = [
                        {
                            "apiVersion": "v1",
                            "kind": "Node",
                            "name": self.node_name,
                            "uid": self.node_uid,
                        }
                    ]
                obj = {
                    "apiVersion": API_VERSION,
                    "kind": "ResourceSlice",
                    "metadata": meta,
                    "spec": {
                        "driver": self.driver_name,
                        "nodeName": self.node_name,
                        "pool": {
                            "name": self.pool_name,
                            "generation": self._generation,
                            "resourceSliceCount": len(chunks),
                        },
                        "devices": chunk,
                    },
                }
                if name in existing:
                    obj["metadata"]["resourceVersion"] = existing[name][
                        "metadata"
                    ]["resourceVersion"]
                    try:
                        out.append(self.client.update_resource_slice(obj))
                    except Conflict:
                        # concurrent writer (e.g. admission rewrite): refetch
                        cur = {
                            s["metadata"]["name"]: s
                            for s in self.client.list_resource_slices(
                                self.driver_name
                            )
                        }[name]
                        obj["metadata"]["resourceVersion"] = cur["metadata"][
                            "resourceVersion"
                        ]
                        out.append(self.client.update_resource_slice(obj))
                else:
                    out.append(self.client.create_resource_slice(obj))
            for name in set(existing) - desired_names:
                self.client.delete_resource_slice(name)
            self._last_fingerprint = fp
            return out

    def unpublish_all(self) -> None:
        """Delete every slice this driver owns on this node (clean shutdown
        parity: reference imex.go:298-316)."""
        with self._lock:
            for s in self.client.list_resource_slices(self.driver_name):
                if s["spec"].get("nodeName") == self.node_name:
                    try:
                        self.client.delete_resource_slice(s["metadata"]["name"])
                    except NotFound:
                        pass
            self._last_fingerprint = None
