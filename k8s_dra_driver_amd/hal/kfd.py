"""KFD-sysfs-only DeviceLib: degraded-mode enumeration without libamd_smi.

Bring-up fallback for nodes where the amd_smi library is absent or
ABI-mismatched (the driver's primary backend links libamd_smi.so.26): the
kernel's KFD topology is always present when amdgpu is loaded, and carries
enough for enumeration + CDI (render minors, gfx arch, CU counts, VRAM
from mem_banks, xGMI adjacency from io_links). Partition CONTROL is not
available here (amd-smi only), so ``set_*_partition`` raise and the
catalog is advertised read-only.

Select with ``--hal kfd``. Never chosen automatically: a node that should
run amdsmi silently degrading would hide breakage (same fail-loud policy
as the amdsmi backend).
"""

from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

from .base import DeviceLib, HalError, HalNotSupported, HalUnavailable
from .model import GpuInfo, XgmiLink
from .sysfs import KfdNode, KfdTopology

log = logging.getLogger(__name__)

#: KFD heap types (kfd_ioctl.h): 1,2 = framebuffer (VRAM) public/private
_HEAP_FB = (1, 2)


class KfdDeviceLib(DeviceLib):
    def __init__(self, sysfs_root: str = "/sys"):
        self.topology = KfdTopology(sysfs_root)
        self._open = False

    # -- lifecycle ---------------------------------------------------------
    def open(self) -> None:
        if not self.topology.available():
            raise HalUnavailable(
                f"KFD topology not present under {self.topology.root} "
                f"(amdgpu/KFD not loaded?)"
            )
        self._open = True

    def close(self) -> None:
        self._open = False

    def _check_open(self) -> None:
        if not self._open:
            raise HalError("device library not open")

    # -- helpers -----------------------------------------------------------
    def _vram_mib(self, node: KfdNode) -> int:
        """Sum framebuffer mem_banks sizes for the node."""
        banks_dir = os.path.join(
            self.topology.nodes_dir, str(node.node_id), "mem_banks"
        )
        total = 0
        try:
            for name in os.listdir(banks_dir):
                props: Dict[str, int] = {}
                try:
                    with open(os.path.join(banks_dir, name, "properties")) as f:
                        for line in f:
                            parts = line.split()
                            if len(parts) == 2 and parts[1].isdigit():
                                props[parts[0]] = int(parts[1])
                except OSError:
                    continue
                if props.get("heap_type") in _HEAP_FB:
                    total += props.get("size_in_bytes", 0)
        except FileNotFoundError:
            pass
        return total // (1024 * 1024)

    # -- enumeration -------------------------------------------------------
    def enumerate(self) -> List[GpuInfo]:
        self._check_open()
        nodes = self.topology.gpu_nodes()
        by_id = {n.node_id: n for n in nodes}
        out: List[GpuInfo] = []
        for index, node in enumerate(nodes):
            uuid = (
                f"kfd-{node.unique_id:016x}"
                if node.unique_id
                else f"kfd-node-{node.node_id}"
            )
            info = GpuInfo(
                index=index,
                uuid=uuid,
                oam_id=index,
                timeslice_effective=False,
                repartition_capable=False,
                product_name="AMD Instinct (KFD)",
                architecture=node.gfx_arch or "unknown",
                pcie_bdf=node.bdf,
                vram_total_mib=self._vram_mib(node),
                cu_count=node.cu_count,
                kfd_node_id=node.node_id,
                render_minor=node.render_minor,
                card_minor=self.topology.card_minor_for_render(
                    node.render_minor
                ),
                compute_partition="SPX",  # control unavailable; see module doc
                memory_partition="NPS1",
                nps_caps=["NPS1"],
                compute_caps=["SPX"],
                xgmi_node_id=node.node_id,
            )
            for peer_id in node.xgmi_peers():
                peer = by_id.get(peer_id)
                if peer is None:
                    continue
                peer_index = nodes.index(peer)
                info.links.append(
                    XgmiLink(
                        peer_oam_id=peer_index,
                        peer_uuid=(
                            f"kfd-{peer.unique_id:016x}"
                            if peer.unique_id
                            else f"kfd-node-{peer.node_id}"
                        ),
                    )
                )
            if info.links:
                info.xgmi_hive_id = "hive-kfd"
            out.append(info)
        return out

    # -- partitioning: read-only in this backend ----------------------------
    def dynamic_repartition_capable(self) -> bool:
        return False  # read-only backend: never advertise carves

    def timeslice_effective(self) -> bool:
        return False  # requests are recorded only in KFD-only mode

    def set_compute_partition(self, gpu_index: int, mode: str) -> None:
        raise HalNotSupported(
            "partition control requires the amdsmi backend (KFD-only mode "
            "is read-only)"
        )

    def set_memory_partition(self, gpu_index: int, mode: str) -> None:
        raise HalNotSupported(
            "partition control requires the amdsmi backend (KFD-only mode "
            "is read-only)"
        )

    # -- scheduler / nodes / health -----------------------------------------
    def set_timeslice_quantum(self, gpu_index: int, quantum_us: Optional[int]) -> None:
        log.info(
            "gpu-%d: time-slice request %s recorded (KFD-only mode)",
            gpu_index,
            quantum_us,
        )

    def device_node_paths(
        self, gpu_index: int, partition_id: Optional[int] = None
    ) -> Dict[str, str]:
        gpus = self.enumerate()
        if gpu_index >= len(gpus):
            raise HalError(f"gpu-{gpu_index} not found")
        g = gpus[gpu_index]
        return {
            "kfd": "/dev/kfd",
            "renderD": f"/dev/dri/renderD{g.render_minor}",
            "card": f"/dev/dri/card{g.card_minor}" if g.card_minor >= 0 else "",
        }

    def health_check(self, gpu_index: int) -> Dict[str, str]:
        gpus = self.enumerate()
        if gpu_index >= len(gpus):
            return {"status": "missing"}
        g = gpus[gpu_index]
        return {"status": "healthy", "uuid": g.uuid, "renderD": str(g.render_minor)}
