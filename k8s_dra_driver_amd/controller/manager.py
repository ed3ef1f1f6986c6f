"""Cluster controller: fabric/node labeling + driver-wide reconciliation.

Reference analog: ``cmd/nvidia-dra-controller`` — whose only job is the
IMEX channel manager (``imex.go``). There is no IMEX on AMD (xGMI is
intra-node, SURVEY.md §0 translation table), so this controller's v0 scope
(SURVEY.md §7 step 8):

- **Node labeling** from published ResourceSlices (the analog of the
  ``nvidia.com/gpu.imex-domain`` label streaming, ``imex.go:206-295``):
  gpu count, product, architecture, xGMI hive, partition modes in use —
  so plain nodeSelectors can target fabric/partition shapes without DRA.
- **Orphan cleanup**: ResourceSlices owned by this driver whose node no
  longer exists are deleted (self-healing the reference gets from the
  vendored controller's informer, resourceslicecontroller.go:407-412).
- Prometheus metrics endpoint parity (``main.go:194-241``).

Level-triggered reconcile loop with a transient-error retry, mirroring the
reference's requeue-after-1-minute behavior (``imex.go:132-151``).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, List, Optional

from .. import DRIVER_NAME
from ..hal.model import DOMAIN
from ..kube.client import KubeClient, NotFound

log = logging.getLogger(__name__)

LABEL_PREFIX = DOMAIN  # gpu.amd.com
RETRY_SECONDS = 60.0


def _attr(device: dict, name: str, default=None):
    basic = device.get("basic", device)
    v = (basic.get("attributes") or {}).get(f"{DOMAIN}/{name}")
    return next(iter(v.values())) if v is not None else default


def labels_for_node(devices: List[dict]) -> Dict[str, str]:
    """Derive node labels from that node's published devices."""
    gpus = [d for d in devices if _attr(d, "type") == "gpu"]
    parts = [d for d in devices if _attr(d, "type") == "partition"]
    # physical GPU count = distinct dies, whether published as a whole
    # GPU, as partitions, or BOTH (prospective-partition mode publishes
    # the whole GPU alongside its would-be partitions)
    dies = {_attr(d, "uuid") for d in gpus} | {
        _attr(p, "parentUUID") for p in parts
    }
    dies.discard(None)
    labels: Dict[str, str] = {
        f"{LABEL_PREFIX}/gpu.present": "true" if devices else "false",
        f"{LABEL_PREFIX}/gpu.count": str(len(dies)),
        f"{LABEL_PREFIX}/device.count": str(len(devices)),
    }
    archs = {_attr(d, "architecture") for d in devices if _attr(d, "architecture")}
    if len(archs) == 1:
        labels[f"{LABEL_PREFIX}/gpu.architecture"] = archs.pop()
    products = {
        str(_attr(d, "productName", "")).replace(" ", "-")
        for d in devices
        if _attr(d, "productName")
    }
    if len(products) == 1:
        labels[f"{LABEL_PREFIX}/gpu.product"] = products.pop()
    hives = {_attr(d, "xgmiHiveId") for d in devices if _attr(d, "xgmiHiveId")}
    if len(hives) == 1:
        labels[f"{LABEL_PREFIX}/xgmi.hive"] = str(hives.pop())
    modes = sorted(
        {
            str(_attr(d, "computePartition"))
            for d in devices
            if _attr(d, "computePartition")
        }
    )
    if modes:
        labels[f"{LABEL_PREFIX}/partition.modes"] = "_".join(modes)
    return labels


class ControllerManager:
    def __init__(
        self,
        kube: KubeClient,
        *,
        poll_interval: float = 10.0,
        manage_labels: bool = True,
        cleanup_orphans: bool = True,
        allocate_claims: bool = False,
    ):
        self.kube = kube
        self.poll_interval = poll_interval
        self.manage_labels = manage_labels
        self.cleanup_orphans = cleanup_orphans
        self.scheduler = None
        if allocate_claims:
            from .scheduler import ClaimScheduler

            self.scheduler = ClaimScheduler(kube)
        self._stop = threading.Event()
        self._kick = threading.Event()  # watch-triggered wakeup
        self._watch = None
        self._thread: Optional[threading.Thread] = None
        self._owned_labels: Dict[str, Dict[str, str]] = {}

    # ------------------------------------------------------------------
    def reconcile_once(self) -> Dict[str, Dict[str, str]]:
        """One level-triggered pass. Returns applied labels per node."""
        if self.scheduler is not None:
            self.scheduler.reconcile_once()
        slices = self.kube.list_resource_slices(DRIVER_NAME)
        by_node: Dict[str, List[dict]] = {}
        for s in slices:
            node = s.get("spec", {}).get("nodeName")
            if node:
                by_node.setdefault(node, []).extend(
                    s["spec"].get("devices") or []
                )

        applied: Dict[str, Dict[str, str]] = {}
        for node, devices in by_node.items():
            try:
                self.kube.get_node(node)
            except NotFound:
                if self.cleanup_orphans:
                    for s in slices:
                        if s["spec"].get("nodeName") == node:
                            log.info(
                                "deleting orphaned ResourceSlice %s (node %s gone)",
                                s["metadata"]["name"],
                                node,
                            )
                            self.kube.delete_resource_slice(s["metadata"]["name"])
                continue
            if not self.manage_labels:
                continue
            labels = labels_for_node(devices)
            prev = self._owned_labels.get(node, {})
            # remove labels we set before that no longer apply
            patch: Dict[str, Optional[str]] = {
                k: None for k in prev if k not in labels
            }
            patch.update(labels)
            if patch != {k: v for k, v in prev.items()}:
                self.kube.patch_node_labels(node, patch)
            self._owned_labels[node] = labels
            applied[node] = labels

        # nodes that lost all slices: clear our labels
        if self.manage_labels:
            for node in list(self._owned_labels):
                if node not in by_node:
                    try:
                        self.kube.patch_node_labels(
                            node,
                            {k: None for k in self._owned_labels[node]},
                        )
                    except NotFound:
                        pass
                    del self._owned_labels[node]
        return applied

    # ------------------------------------------------------------------
    def run(self) -> None:
        """Level-triggered reconcile loop: wakes on watch events (claim
        churn) or the resync interval; transient-error retry parity with
        the reference (imex.go:132-151)."""
        backoff = self.poll_interval
        while not self._stop.is_set():
            try:
                self.reconcile_once()
                backoff = self.poll_interval
            except Exception:
                log.exception("reconcile failed; retrying in %.0fs", RETRY_SECONDS)
                backoff = RETRY_SECONDS
            self._kick.wait(backoff)
            self._kick.clear()

    def start(self) -> None:
        if self.scheduler is not None:
            # informer-style wakeup: allocate new claims promptly instead
            # of waiting out the poll interval (client-go informer analog)
            def on_claim(event_type: str, obj: dict) -> None:
                if event_type in ("ADDED", "MODIFIED") and not (
                    obj.get("status") or {}
                ).get("allocation"):
                    self._kick.set()

            try:
                self._watch = self.kube.watch_resource_claims(on_claim)
            except Exception:
                log.exception("claim watch unavailable; polling only")
        self._thread = threading.Thread(
            target=self.run, name="controller-reconcile", daemon=True
        )
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        self._kick.set()
        if self._watch is not None:
            self._watch.stop()
        if self._thread:
            self._thread.join(timeout=5)
