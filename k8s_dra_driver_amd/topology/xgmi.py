"""xGMI topology model and placement scoring.

MI355X xGMI is point-to-point: 7 links x ~153 GB/s per GPU on an 8-GPU OAM
baseboard, no switch. Ring collectives are per-link bound,
so the quality of a multi-GPU claim placement = how many of the claim's
GPUs are mutual xGMI neighbors (and, for partitions, whether they share a
parent die / memory domain).

The scorer consumes published ResourceSlice Device dicts (the same
attributes CEL sees), so the allocator, controller and tests all reason
over one representation.
"""

from __future__ import annotations

import itertools
from typing import Dict, List, Sequence

from ..hal.model import DOMAIN


def _attr(device: dict, name: str, default=None):
    basic = device.get("basic", device)
    v = (basic.get("attributes") or {}).get(f"{DOMAIN}/{name}")
    if v is None:
        return default
    return next(iter(v.values()))


def peer_oams(device: dict) -> set:
    raw = _attr(device, "xgmiPeerOamIds", "") or ""
    return {int(x) for x in raw.split(",") if x.strip().isdigit()}


def pair_score(a: dict, b: dict) -> int:
    """Connectivity score of one device pair (higher = better placement).

    - partitions of the same parent die: 100 (share L2-adjacent fabric
      and memory domains; the parentUUID constraint case of gpu-test4)
    - same parent + same memory domain: +20
    - direct xGMI neighbors (mutual): 10
    - same hive but not direct neighbors: 1
    - unrelated: 0
    """
    pa, pb = _attr(a, "parentUUID"), _attr(b, "parentUUID")
    if pa is not None and pa == pb:
        da, db = _attr(a, "memoryDomain"), _attr(b, "memoryDomain")
        return 120 if (da is not None and da == db) else 100
    oa, ob = _attr(a, "oamId"), _attr(b, "oamId")
    if oa is not None and ob is not None and oa != ob:
        if ob in peer_oams(a) and oa in peer_oams(b):
            return 10
    ha, hb = _attr(a, "xgmiHiveId"), _attr(b, "xgmiHiveId")
    if ha and ha == hb:
        return 1
    return 0


def subset_score(devices: Sequence[dict]) -> int:
    """Sum of pairwise scores — what the allocator maximizes."""
    return sum(
        pair_score(a, b) for a, b in itertools.combinations(devices, 2)
    )


def is_fully_connected(devices: Sequence[dict]) -> bool:
    """Every pair is xGMI-adjacent (or same-parent partitions)."""
    return all(
        pair_score(a, b) >= 10
        for a, b in itertools.combinations(devices, 2)
    )


def pick_best_subset(
    candidates: List[dict], count: int, *, exhaustive_limit: int = 16
) -> List[dict]:
    """Best-scoring subset of ``count`` devices.

    Exhaustive for small candidate sets (C(16,k) is fine), greedy beyond:
    seed with the device of highest total affinity, then grow by best
    marginal score.
    """
    if count >= len(candidates):
        return list(candidates)
    if len(candidates) <= exhaustive_limit:
        best, best_score = None, -1
        for combo in itertools.combinations(candidates, count):
            s = subset_score(combo)
            if s > best_score:
                best, best_score = combo, s
        return list(best or candidates[:count])
    # greedy
    totals = [
        (sum(pair_score(c, o) for o in candidates if o is not c), i)
        for i, c in enumerate(candidates)
    ]
    totals.sort(reverse=True)
    chosen = [candidates[totals[0][1]]]
    remaining = [c for c in candidates if c is not chosen[0]]
    while len(chosen) < count:
        best_i, best_s = 0, -1
        for i, c in enumerate(remaining):
            s = sum(pair_score(c, o) for o in chosen)
            if s > best_s:
                best_i, best_s = i, s
        chosen.append(remaining.pop(best_i))
    return chosen


def node_fabric_summary(devices: Sequence[dict]) -> Dict[str, object]:
    """Controller-side summary for node labeling: hive ids + link counts."""
    hives = {}
    for d in devices:
        h = _attr(d, "xgmiHiveId")
        if h:
            hives.setdefault(h, 0)
            hives[h] += 1
    return {
        "hives": hives,
        "fully_connected": is_fully_connected(
            [d for d in devices if _attr(d, "type") == "gpu"]
        )
        if devices
        else False,
    }
