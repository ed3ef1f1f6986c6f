"""Topology-aware claim scheduler (controller mode).

The north-star design gives the cluster controller an xGMI-topology-aware
allocator (BASELINE.json north_star). With DRA structured parameters the
default kube-scheduler *can* allocate, but it picks arbitrarily among
feasible device sets; this controller loop allocates pending ResourceClaims
itself using :mod:`k8s_dra_driver_amd.allocator` — CEL selectors,
matchAttribute constraints, and placement scoring that maximizes mutual
xGMI adjacency (and same-die co-location for partitions).

Per pass (level-triggered, like the label reconciler):

1. gather published devices per node from this driver's ResourceSlices;
2. compute per-node in-use sets from already-allocated claims;
3. for each unallocated claim whose requests target our DeviceClasses,
   try every candidate node, keep the highest-scoring feasible placement,
   and write ``status.allocation`` (with a node selector).

Enable with ``--allocate-claims`` (off by default: in clusters where
kube-scheduler performs DRA allocation, two allocators would race).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Set, Tuple

from .. import DRIVER_NAME
from ..allocator.structured import (
    AllocationError,
    Allocator,
    DeviceClass,
)
from ..topology.xgmi import subset_score
from ..kube.client import KubeClient

log = logging.getLogger(__name__)


class ClaimScheduler:
    def __init__(self, kube: KubeClient, *, driver: str = DRIVER_NAME):
        self.kube = kube
        self.driver = driver

    # ------------------------------------------------------------------
    def _allocator(self) -> Allocator:
        classes = {}
        try:
            for obj in self.kube.get_device_classes():
                dc = DeviceClass.from_obj(obj)
                classes[dc.name] = dc
        except NotImplementedError:
            pass
        return Allocator(classes, driver=self.driver)

    def _devices_by_node(self) -> Dict[str, Tuple[str, List[dict]]]:
        """node -> (pool name, devices)."""
        out: Dict[str, Tuple[str, List[dict]]] = {}
        for s in self.kube.list_resource_slices(self.driver):
            spec = s.get("spec", {})
            node = spec.get("nodeName")
            if not node:
                continue
            pool = (spec.get("pool") or {}).get("name", node)
            cur = out.setdefault(node, (pool, []))
            cur[1].extend(spec.get("devices") or [])
        return out

    def _in_use_by_pool(self, claims: List[dict]) -> Dict[str, Set[str]]:
        out: Dict[str, Set[str]] = {}
        for c in claims:
            alloc = (c.get("status") or {}).get("allocation")
            if not alloc:
                continue
            for r in (alloc.get("devices") or {}).get("results") or []:
                if r.get("driver") != self.driver:
                    continue
                if r.get("adminAccess"):
                    continue  # monitoring claims don't consume devices
                out.setdefault(r.get("pool", ""), set()).add(r.get("device"))
        return out

    def _targets_our_classes(self, claim: dict, alloc: Allocator) -> bool:
        reqs = (
            (claim.get("spec") or {}).get("devices") or {}
        ).get("requests") or []
        return bool(reqs) and all(
            r.get("deviceClassName") in alloc.classes for r in reqs
        )

    # ------------------------------------------------------------------
    def reconcile_once(self) -> List[str]:
        """Allocate pending claims; returns the UIDs allocated this pass."""
        try:
            claims = self.kube.list_resource_claims()
        except NotImplementedError:
            return []
        alloc = self._allocator()
        nodes = self._devices_by_node()
        if not nodes:
            return []
        in_use = self._in_use_by_pool(claims)
        done: List[str] = []
        for claim in claims:
            if (claim.get("status") or {}).get("allocation"):
                continue
            if not self._targets_our_classes(claim, alloc):
                continue
            placed = self._place(claim, alloc, nodes, in_use)
            if placed is None:
                continue
            node, results = placed
            pool = nodes[node][0]
            try:
                self.kube.update_resource_claim_status(claim)
            except Exception:
                log.exception(
                    "allocation status write failed for %s/%s",
                    claim["metadata"].get("namespace"),
                    claim["metadata"].get("name"),
                )
                continue
            for r in results:
                in_use.setdefault(pool, set()).add(r.device)
            done.append(claim["metadata"].get("uid", ""))
        if done:
            log.info("allocated %d claim(s)", len(done))
        return done

    def _place(
        self,
        claim: dict,
        alloc: Allocator,
        nodes: Dict[str, Tuple[str, List[dict]]],
        in_use: Dict[str, Set[str]],
    ) -> Optional[Tuple[str, list]]:
        """Best node for the claim by topology score; mutates the claim's
        status on success (allocate_into_claim)."""
        best = None  # (score, node, results, allocated_claim_status)
        for node, (pool, devices) in nodes.items():
            try:
                results = alloc.allocate(
                    claim.get("spec", {}),
                    devices,
                    pool=pool,
                    in_use=in_use.get(pool, set()),
                )
            except AllocationError:
                continue
            by_name = {d["name"]: d for d in devices}
            score = subset_score([by_name[r.device] for r in results])
            if best is None or score > best[0]:
                best = (score, node, results)
        if best is None:
            return None
        _, node, results = best
        alloc.attach_allocation(claim, results, node_name=node)
        return node, results
