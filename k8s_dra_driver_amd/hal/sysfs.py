"""KFD / DRM sysfs topology parsing.

The ``/proc/devices``-parsing analog of the reference (``nvlib.go:446-488``)
— but richer: KFD exposes the full node topology under
``/sys/class/kfd/kfd/topology/nodes/<n>/``:

- ``properties``: key/value lines (simd_count, gfx_target_version,
  drm_render_minor, location_id, unique_id, ...). Each *compute partition*
  appears as its own KFD node with its own drm_render_minor — this is how
  partition device nodes are discovered after a mode switch.
- ``io_links/<i>/properties``: inter-node links (type 11 = xGMI) giving the
  fabric adjacency the topology-aware allocator publishes.

Root is injectable so tests run against a fixture tree.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List

KFD_IOLINK_TYPE_PCIE = 2
KFD_IOLINK_TYPE_XGMI = 11


@dataclass
class KfdIoLink:
    node_from: int
    node_to: int
    link_type: int
    weight: int = 0
    min_bandwidth: int = 0
    max_bandwidth: int = 0

    @property
    def is_xgmi(self) -> bool:
        return self.link_type == KFD_IOLINK_TYPE_XGMI


@dataclass
class KfdNode:
    node_id: int
    properties: Dict[str, int] = field(default_factory=dict)
    io_links: List[KfdIoLink] = field(default_factory=list)

    @property
    def is_gpu(self) -> bool:
        return self.properties.get("simd_count", 0) > 0

    @property
    def render_minor(self) -> int:
        return self.properties.get("drm_render_minor", -1)

    @property
    def gfx_target_version(self) -> int:
        return self.properties.get("gfx_target_version", 0)

    @property
    def gfx_arch(self) -> str:
        """Decode gfx_target_version (major*10000+minor*100+step) to the
        gfx name, e.g. 90500 -> gfx950."""
        v = self.gfx_target_version
        if v == 0:
            return ""
        major, minor, step = v // 10000, (v // 100) % 100, v % 100
        return f"gfx{major}{minor:x}{step:x}"

    @property
    def cu_count(self) -> int:
        simd = self.properties.get("simd_count", 0)
        per_cu = self.properties.get("simd_per_cu", 4) or 4
        return simd // per_cu

    @property
    def unique_id(self) -> int:
        return self.properties.get("unique_id", 0)

    @property
    def location_id(self) -> int:
        return self.properties.get("location_id", 0)

    @property
    def domain(self) -> int:
        return self.properties.get("domain", 0)

    @property
    def bdf(self) -> str:
        loc = self.location_id
        return f"{self.domain:04x}:{(loc >> 8) & 0xFF:02x}:{(loc >> 3) & 0x1F:02x}.{loc & 0x7}"

    def xgmi_peers(self) -> List[int]:
        return sorted(
            l.node_to for l in self.io_links if l.is_xgmi and l.node_to != self.node_id
        )


def _parse_properties(path: str) -> Dict[str, int]:
    out: Dict[str, int] = {}
    try:
        with open(path) as f:
            for line in f:
                parts = line.split()
                if len(parts) == 2:
                    try:
                        out[parts[0]] = int(parts[1])
                    except ValueError:
                        pass
    except (FileNotFoundError, PermissionError):
        pass
    return out


class KfdTopology:
    def __init__(self, sysfs_root: str = "/sys"):
        self.root = sysfs_root
        self.nodes_dir = os.path.join(
            sysfs_root, "class", "kfd", "kfd", "topology", "nodes"
        )

    def available(self) -> bool:
        return os.path.isdir(self.nodes_dir)

    def nodes(self) -> List[KfdNode]:
        out: List[KfdNode] = []
        if not self.available():
            return out
        for name in sorted(os.listdir(self.nodes_dir), key=lambda s: int(s) if s.isdigit() else -1):
            if not name.isdigit():
                continue
            node_dir = os.path.join(self.nodes_dir, name)
            node = KfdNode(
                node_id=int(name),
                properties=_parse_properties(os.path.join(node_dir, "properties")),
            )
            links_dir = os.path.join(node_dir, "io_links")
            if os.path.isdir(links_dir):
                for ln in sorted(os.listdir(links_dir)):
                    props = _parse_properties(
                        os.path.join(links_dir, ln, "properties")
                    )
                    if props:
                        node.io_links.append(
                            KfdIoLink(
                                node_from=props.get("node_from", node.node_id),
                                node_to=props.get("node_to", -1),
                                link_type=props.get("type", -1),
                                weight=props.get("weight", 0),
                                min_bandwidth=props.get("min_bandwidth", 0),
                                max_bandwidth=props.get("max_bandwidth", 0),
                            )
                        )
            out.append(node)
        return out

    def gpu_nodes(self) -> List[KfdNode]:
        return [n for n in self.nodes() if n.is_gpu]

    def card_minor_for_render(self, render_minor: int) -> int:
        """Map renderD minor -> cardN via /sys/class/drm (same PCI device)."""
        drm = os.path.join(self.root, "class", "drm")
        try:
            target = os.path.realpath(
                os.path.join(drm, f"renderD{render_minor}", "device")
            )
            for name in os.listdir(drm):
                if name.startswith("card") and name[4:].isdigit():
                    dev = os.path.realpath(os.path.join(drm, name, "device"))
                    if dev == target:
                        return int(name[4:])
        except (FileNotFoundError, PermissionError):
            pass
        return -1
