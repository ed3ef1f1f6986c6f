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
  slices and creates/updates/deletes to match;
- **drift self-healing** (``resourceslicecontroller.go:407-431``): with
  :meth:`start_self_heal` the publisher watches this driver's slices and
  re-publishes whenever one is externally deleted or mutated — the
  fingerprint short-circuit is bypassed when observed state diverges from
  desired, so a slice killed out from under the plugin comes back without
  waiting for the next repartition/health event.
"""

from __future__ import annotations

import hashlib
import json
import logging
import threading
from typing import Dict, List, Optional

from .client import Conflict, KubeClient, NotFound

log = logging.getLogger(__name__)

MAX_DEVICES_PER_SLICE = 128
API_VERSION = "resource.k8s.io/v1beta1"
#: versions this publisher can emit, most-preferred first (v1 GA in
#: K8s 1.34 and v1beta2 share the flattened Device shape)
SUPPORTED_VERSIONS = ("v1", "v1beta2", "v1beta1")


def _devices_fingerprint(devices: List[dict]) -> str:
    data = json.dumps(devices, sort_keys=True).encode()
    return hashlib.sha256(data).hexdigest()[:16]


def flatten_device_v1beta2(dev: dict) -> dict:
    """v1beta1 Device ({name, basic:{attributes, capacity}}) -> the
    flattened v1beta2/v1 shape (K8s 1.33+): attributes/capacity move to
    the top level (reference tracks this as the draplugin.go:342-350
    multi-version pattern)."""
    basic = dev.get("basic")
    if basic is None:
        return dev  # already flat
    out: dict = {"name": dev["name"]}
    out["attributes"] = basic.get("attributes", {})
    out["capacity"] = basic.get("capacity", {})
    for k, v in basic.items():
        if k not in ("attributes", "capacity"):
            out[k] = v
    for k, v in dev.items():
        if k not in ("name", "basic"):
            out[k] = v
    return out


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
        # RLock: a synchronous watch client (InMemoryKube) delivers the
        # events of our own heal writes back into _on_slice_event on the
        # same thread while _publish_locked still holds the lock.
        self._lock = threading.RLock()
        self._generation = 0
        self._last_fingerprint: Optional[str] = None
        #: desired spec per slice name, cached for drift detection
        self._desired_specs: Dict[str, dict] = {}
        #: the raw publish() inputs backing the current desired state
        self._last_publish: Optional[tuple] = None
        self._shared_counters: List[dict] = []
        self._warned_counters = False
        self._watch = None
        #: observability: count of heals performed (tests/metrics)
        self.heal_count = 0
        #: optional callback per heal (metrics wiring)
        self.on_heal = None
        #: negotiated resource.k8s.io version (resolved on first publish)
        self._api_version: Optional[str] = None

    def negotiated_version(self) -> str:
        """Public accessor (triggers negotiation): lets callers adapt
        behavior to the API generation — e.g. taint unhealthy devices on
        v1beta2+ instead of unpublishing them."""
        with self._lock:
            return self._negotiate_version()

    def _negotiate_version(self) -> str:
        """Pick the newest mutually-supported resource.k8s.io version
        (single code path: devices are built v1beta1-shaped and flattened
        when the apiserver prefers v1beta2)."""
        if self._api_version is None:
            try:
                served = self.client.resource_api_versions()
            except Exception:
                served = ["v1beta1"]
            self._api_version = next(
                (v for v in SUPPORTED_VERSIONS if v in served), "v1beta1"
            )
            log.info(
                "publishing ResourceSlices as resource.k8s.io/%s "
                "(served: %s)",
                self._api_version,
                served,
            )
        return self._api_version

    def _slice_name(self, index: int) -> str:
        safe_driver = self.driver_name.replace("/", "-").replace(".", "-")
        return f"{self.node_name}-{safe_driver}-{index}"

    # ------------------------------------------------------------------
    # publication
    # ------------------------------------------------------------------
    def publish(
        self,
        devices: List[dict],
        *,
        counter_devices: Optional[List[dict]] = None,
        shared_counters: Optional[List[dict]] = None,
    ) -> List[dict]:
        """Reconcile the published slices to carry exactly ``devices``.

        ``counter_devices`` + ``shared_counters`` express DRA
        partitionable-devices semantics (K8s 1.33, sharedCounters /
        consumesCounters): prospective partition devices that overlap the
        whole GPU via a per-GPU counter set. They are published only when
        the negotiated API version supports counters (v1beta2/v1); on
        v1beta1 they are dropped (logged once) and any consumesCounters
        keys on ``devices`` are stripped — the scheduler there cannot see
        the overlap, so offering both shapes would double-allocate.

        Returns the slice objects as stored. No-op (no API calls beyond the
        LIST) when the device set is unchanged AND the observed slices match
        the desired state — cheap to call after every enumeration, but never
        blind to external tampering.
        """
        with self._lock:
            self._last_publish = (
                list(devices),
                list(counter_devices or []),
                list(shared_counters or []),
            )
            return self._publish_locked(
                devices,
                counter_devices=counter_devices,
                shared_counters=shared_counters,
            )

    def _observed_matches_desired(self, existing: Dict[str, dict]) -> bool:
        if set(existing) != set(self._desired_specs):
            return False
        for name, want_spec in self._desired_specs.items():
            if existing[name].get("spec") != want_spec:
                return False
        return True

    def _publish_locked(
        self,
        devices: List[dict],
        *,
        counter_devices: Optional[List[dict]] = None,
        shared_counters: Optional[List[dict]] = None,
    ) -> List[dict]:
        version = self._negotiate_version()
        counters_ok = version != "v1beta1"
        if counters_ok:
            devices = [flatten_device_v1beta2(d) for d in devices]
            if counter_devices:
                devices = devices + [
                    flatten_device_v1beta2(d) for d in counter_devices
                ]
            self._shared_counters = list(shared_counters or [])
        else:
            if counter_devices and not self._warned_counters:
                self._warned_counters = True
                log.warning(
                    "apiserver serves only v1beta1 (no sharedCounters): "
                    "dropping %d prospective partition device(s); "
                    "scheduler-driven dynamic partitioning needs "
                    "resource.k8s.io v1beta2+ (K8s 1.33)",
                    len(counter_devices),
                )
            devices = [
                {k: v for k, v in d.items() if k != "consumesCounters"}
                for d in devices
            ]
            self._shared_counters = []
        fp = _devices_fingerprint(devices + self._shared_counters)
        existing = {
            s["metadata"]["name"]: s
            for s in self.client.list_resource_slices(self.driver_name)
            if s["spec"].get("nodeName") == self.node_name
        }
        if (
            fp == self._last_fingerprint
            and existing
            and self._observed_matches_desired(existing)
        ):
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
        self._desired_specs = {}
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
            spec = {
                "driver": self.driver_name,
                "nodeName": self.node_name,
                "pool": {
                    "name": self.pool_name,
                    "generation": self._generation,
                    "resourceSliceCount": len(chunks),
                },
                "devices": chunk,
            }
            if self._shared_counters:
                # counter sets ride on every slice of the pool so any
                # slice alone carries the full overlap model
                spec["sharedCounters"] = list(self._shared_counters)
            obj = {
                "apiVersion": f"resource.k8s.io/{version}",
                "kind": "ResourceSlice",
                "metadata": meta,
                "spec": spec,
            }
            # register desired BEFORE the write: a synchronous watcher
            # delivers our own create/update event re-entrantly, and it
            # must compare against the new spec (match -> no heal loop)
            self._desired_specs[name] = spec
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
                try:
                    out.append(self.client.create_resource_slice(obj))
                except Conflict:
                    # raced an external re-creation: overwrite it
                    cur = {
                        s["metadata"]["name"]: s
                        for s in self.client.list_resource_slices(
                            self.driver_name
                        )
                    }.get(name)
                    if cur is None:
                        raise
                    obj["metadata"]["resourceVersion"] = cur["metadata"][
                        "resourceVersion"
                    ]
                    out.append(self.client.update_resource_slice(obj))
        for name in set(existing) - desired_names:
            self.client.delete_resource_slice(name)
        self._last_fingerprint = fp
        return out

    # ------------------------------------------------------------------
    # drift self-heal (resourceslicecontroller.go:407-431 parity)
    # ------------------------------------------------------------------
    def start_self_heal(self) -> bool:
        """Watch this driver's slices and republish on external drift.

        Returns False when the client cannot watch (callers may poll)."""
        if self._watch is not None:
            return True
        w = self.client.watch_resource_slices(self._on_slice_event)
        if w is None:
            return False
        self._watch = w
        return True

    def stop_self_heal(self) -> None:
        if self._watch is not None:
            self._watch.stop()
            self._watch = None

    def _on_slice_event(self, etype: str, obj: dict) -> None:
        try:
            spec = obj.get("spec") or {}
            name = (obj.get("metadata") or {}).get("name", "")
            with self._lock:
                if self._last_publish is None or not self._desired_specs:
                    return  # nothing published yet
                if name not in self._desired_specs:
                    # not ours (or an orphan being deleted) — ignore
                    return
                if etype == "DELETED":
                    drifted = True
                else:
                    drifted = spec != self._desired_specs[name]
                if not drifted:
                    return
                log.warning(
                    "resourceslice %s drifted externally (%s); re-publishing",
                    name,
                    etype,
                )
                self.heal_count += 1
                if self.on_heal is not None:
                    try:
                        self.on_heal()
                    except Exception:
                        log.debug("heal callback failed", exc_info=True)
                devs, cdevs, scnt = self._last_publish
                self._publish_locked(
                    list(devs),
                    counter_devices=list(cdevs),
                    shared_counters=list(scnt),
                )
        except Exception:
            log.exception("slice self-heal failed for event %s", etype)

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
            self._desired_specs = {}
            self._last_publish = None
