"""amdsmi-backed DeviceLib: the real-hardware HAL for MI355X nodes.

Combines two native sources (the go-nvml + /proc parsing analog,
``nvlib.go:59-63,446-488``):

- the in-tree ``_amdhal`` C++ extension (libamd_smi): identity, VRAM,
  driver, partition modes/caps, xGMI hive and link metrics;
- KFD sysfs topology (:mod:`.sysfs`): render minors, gfx arch, CU counts,
  per-partition KFD nodes and xGMI adjacency.

Grouping model: AMD SMI reports one *processor* per KFD device — so a GPU
in CPX mode shows up as 8 processors sharing an OAM id. ``enumerate()``
groups processors into physical GPUs and attaches PartitionedDeviceInfo
entries for non-SPX modes, which is exactly the shape the rest of the
driver consumes from the fake backend too.

This backend FAILS LOUDLY when the native extension or the amdgpu driver
is missing — a GPU node silently falling back to fake hardware would
advertise devices that cannot run anything.
"""

from __future__ import annotations

import logging
import os
from collections import defaultdict
from typing import Dict, List, Optional

from ..partition.catalog import COMPUTE_MODES, DEFAULT_VALID_NPS, make_profile
from .base import DeviceLib, HalError, HalUnavailable
from .model import GpuInfo, PartitionedDeviceInfo, XgmiLink
from .sysfs import KfdTopology

log = logging.getLogger(__name__)

_VRAM_TYPES = {
    1: "HBM",
    2: "HBM2",
    3: "HBM2E",
    4: "HBM3",
    5: "HBM3E",
}


def _rocm_version() -> str:
    for root in (os.environ.get("ROCM_PATH", "/opt/rocm"),):
        try:
            with open(os.path.join(root, ".info", "version")) as f:
                return f.read().strip()
        except OSError:
            continue
    return ""


class AmdSmiDeviceLib(DeviceLib):
    def __init__(self, sysfs_root: str = "/sys", *, ext=None):
        """``ext``: test seam — an object with the _amdhal surface; the
        real native extension is imported when omitted."""
        self._ext = None
        self._ext_override = ext
        self.topology = KfdTopology(sysfs_root)
        self._timeslice: Dict[int, Optional[int]] = {}
        self._quantum_knob_writable: Optional[bool] = None
        self._quantum_default_ms: int = 0
        #: learned capability: flips False after a failed partition set
        self._repartition_capable: Optional[bool] = None
        self._rocm = _rocm_version()

    # -- lifecycle ---------------------------------------------------------
    def open(self) -> None:
        if self._ext is not None:
            return
        if self._ext_override is not None:
            self._ext_override.init()
            self._ext = self._ext_override
            return
        try:
            from .. import _amdhal  # in-tree native extension
        except ImportError as e:
            raise HalUnavailable(
                "native extension k8s_dra_driver_amd._amdhal is not built "
                "(run `python setup.py build_ext --inplace`); refusing to "
                f"run on GPU hardware without it: {e}"
            ) from e
        try:
            _amdhal.init()
        except RuntimeError as e:
            raise HalUnavailable(f"amdsmi_init failed: {e}") from e
        self._ext = _amdhal

    def close(self) -> None:
        if self._ext is not None:
            self._ext.shutdown()
            self._ext = None

    def _require(self):
        if self._ext is None:
            raise HalError("device library not open")
        return self._ext

    # -- enumeration -------------------------------------------------------
    def enumerate(self) -> List[GpuInfo]:
        ext = self._require()
        procs = ext.enumerate()
        kfd_by_node = {n.node_id: n for n in self.topology.gpu_nodes()}

        # group processors into physical GPUs
        groups: Dict[object, List[dict]] = defaultdict(list)
        for p in procs:
            oam = p.get("oam_id", 0xFFFFFFFF)
            key = oam if oam != 0xFFFFFFFF else p.get("uuid", p["index"])
            groups[key].append(p)

        out: List[GpuInfo] = []
        # Order groups numerically first (int OAM ids), then lexically for
        # string keys (UUID fallback), so gpu_index is stable across
        # restarts on nodes with >=10 devices ('10' must not sort before
        # '2' — indices are persisted in checkpoints as parent_gpu_index
        # and drive holder/drain/partition bookkeeping).
        for gpu_index, (key, members) in enumerate(
            sorted(
                groups.items(),
                key=lambda kv: (isinstance(kv[0], str), kv[0]),
            )
        ):
            members.sort(key=lambda p: p.get("current_partition_id", 0))
            head = members[0]
            compute_mode = head.get("compute_partition", "SPX") or "SPX"
            memory_mode = head.get("memory_partition", "NPS1") or "NPS1"
            kfd_node = kfd_by_node.get(head.get("kfd_node_id", -1))

            # Assumption (pinned in tests/test_amdsmi_hal.py): in non-SPX
            # modes each partition processor reports its own VRAM share,
            # so the physical total is the sum over members. Unverifiable
            # on the round-1 pool (partition switching blocked); revisit
            # against bare metal.
            vram_mb = int(head.get("vram_size_mb", 0))
            cu_total = sum(
                int(m.get("num_compute_units", 0) or 0) for m in members
            )
            arch = kfd_node.gfx_arch if kfd_node else ""
            gfx_ver = head.get("target_graphics_version", 0)
            if not arch and gfx_ver:
                v = int(gfx_ver)
                arch = f"gfx{v // 10000}{(v // 100) % 100:x}{v % 100:x}"

            info = GpuInfo(
                index=gpu_index,
                uuid=str(head.get("uuid", f"amd-gpu-{gpu_index}")),
                oam_id=int(head.get("oam_id", gpu_index) or gpu_index),
                product_name=str(
                    head.get("product_name")
                    or head.get("market_name", "AMD Instinct")
                ),
                architecture=arch or "unknown",
                device_id=int(head.get("device_id", 0)),
                pcie_bdf=str(head.get("bdf", "")),
                vram_total_mib=vram_mb * len(members)
                if compute_mode != "SPX" and vram_mb
                else vram_mb,
                vram_type=_VRAM_TYPES.get(int(head.get("vram_type", 0)), "VRAM"),
                cu_count=cu_total or (kfd_node.cu_count if kfd_node else 0),
                xcd_count=8,
                driver_version=str(head.get("driver_version", "")),
                rocm_version=self._rocm,
                kfd_node_id=int(head.get("kfd_node_id", -1)),
                render_minor=kfd_node.render_minor if kfd_node else -1,
                card_minor=self.topology.card_minor_for_render(
                    kfd_node.render_minor
                )
                if kfd_node
                else -1,
                compute_partition=compute_mode,
                memory_partition=memory_mode,
                nps_caps=list(head.get("nps_caps", ["NPS1"])),
                compute_caps=list(COMPUTE_MODES),
                xgmi_hive_id=f"hive-{head.get('xgmi_hive_id', 0):#x}"
                if head.get("xgmi_hive_id")
                else "",
                xgmi_node_id=int(head.get("xgmi_node_id", -1)),
                timeslice_effective=self.timeslice_effective(),
            )
            # capability probe (no recursion: uses the card minor in hand)
            if self._repartition_capable is None and info.card_minor >= 0:
                self._repartition_capable = self._probe_repartition_path(
                    info.card_minor
                )
            info.repartition_capable = bool(self._repartition_capable)

            if compute_mode != "SPX" and len(members) > 1:
                try:
                    prof = make_profile(
                        compute_mode,
                        memory_mode if memory_mode in DEFAULT_VALID_NPS.get(compute_mode, ()) else "NPS1",
                        vram_total_mib=info.vram_total_mib or 288 * 1024,
                        cu_count=info.cu_count or 256,
                    )
                except ValueError:
                    prof = make_profile(compute_mode, "NPS1")
                for m in members:
                    pid = int(m.get("current_partition_id", 0))
                    knode = kfd_by_node.get(m.get("kfd_node_id", -1))
                    info.partitions.append(
                        PartitionedDeviceInfo(
                            parent_index=gpu_index,
                            parent_uuid=info.uuid,
                            partition_id=pid,
                            profile=prof,
                            kfd_node_id=int(m.get("kfd_node_id", -1)),
                            render_minor=knode.render_minor if knode else -1,
                            card_minor=self.topology.card_minor_for_render(
                                knode.render_minor
                            )
                            if knode
                            else -1,
                        )
                    )
            out.append(info)

        self._attach_links(out, procs)
        return out

    def _attach_links(self, gpus: List[GpuInfo], procs: List[dict]) -> None:
        """xGMI adjacency from amdsmi link metrics (bdf-keyed)."""
        by_bdf = {g.pcie_bdf: g for g in gpus if g.pcie_bdf}
        for g in gpus:
            head = next(
                (p for p in procs if str(p.get("bdf", "")) == g.pcie_bdf), None
            )
            if head is None:
                continue
            for l in head.get("links", []):
                if l.get("link_type") != "XGMI":
                    continue
                peer = by_bdf.get(l.get("bdf", ""))
                if peer is None or peer.index == g.index:
                    continue
                g.links.append(
                    XgmiLink(
                        peer_oam_id=peer.oam_id,
                        peer_uuid=peer.uuid,
                        max_bandwidth_gbps=int(l.get("max_bandwidth_gbs", 0))
                        or 153,
                    )
                )

    # -- partitioning ------------------------------------------------------
    def dynamic_repartition_capable(self) -> bool:
        """Best-effort probe: can this node actually switch partition
        modes? Checks the bare-metal sysfs control is present and writable
        (virtualized/shared pools expose it read-only or reject writes —
        round-1 saw AMDSMI_STATUS_UNKNOWN_ERROR there), and
        learns from failed sets. Never raises."""
        if self._repartition_capable is not None:
            return self._repartition_capable
        capable = False
        try:
            for g in self.enumerate():  # enumerate() fills the cache
                if g.card_minor >= 0:
                    capable = self._probe_repartition_path(g.card_minor)
                    break
        except Exception:
            capable = False
        self._repartition_capable = capable
        return capable

    def _probe_repartition_path(self, card_minor: int) -> bool:
        path = (
            f"{self.topology.root}/class/drm/card{card_minor}"
            "/device/current_compute_partition"
        )
        return os.path.isfile(path) and os.access(path, os.W_OK)

    def set_compute_partition(self, gpu_index: int, mode: str) -> None:
        ext = self._require()
        proc_index = self._head_proc_index(gpu_index)
        try:
            ext.set_compute_partition(proc_index, mode)
        except Exception:
            self._repartition_capable = False  # learned: this box refuses
            raise
        self._repartition_capable = True
        ext.reinit()  # processor handles change with the KFD device set

    def set_memory_partition(self, gpu_index: int, mode: str) -> None:
        ext = self._require()
        proc_index = self._head_proc_index(gpu_index)
        try:
            ext.set_memory_partition(proc_index, mode)
        except Exception:
            self._repartition_capable = False
            raise
        self._repartition_capable = True
        ext.reinit()

    def _head_proc_index(self, gpu_index: int) -> int:
        """amdsmi processor index of the GPU's partition-0 processor."""
        ext = self._require()
        procs = ext.enumerate()
        groups: Dict[object, List[dict]] = defaultdict(list)
        for p in procs:
            oam = p.get("oam_id", 0xFFFFFFFF)
            key = oam if oam != 0xFFFFFFFF else p.get("uuid", p["index"])
            groups[key].append(p)
        ordered = sorted(
            groups.items(), key=lambda kv: (isinstance(kv[0], str), kv[0])
        )
        if gpu_index >= len(ordered):
            raise HalError(f"gpu-{gpu_index} not found ({len(ordered)} GPUs)")
        members = sorted(
            ordered[gpu_index][1],
            key=lambda p: p.get("current_partition_id", 0),
        )
        return int(members[0]["index"])

    # -- scheduler controls --------------------------------------------------
    #: runtime-writable amdgpu module param controlling how long a compute
    #: queue may run before the HWS preempts it — the closest real analog
    #: of the reference's per-GPU compute-policy time-slice. Global (all
    #: GPUs on the node), probed once at first use.
    QUEUE_PREEMPTION_PARAM = (
        "/sys/module/amdgpu/parameters/queue_preemption_timeout_ms"
    )

    def _probe_quantum_knob(self) -> bool:
        """True when the node exposes a writable scheduler-quantum knob."""
        if self._quantum_knob_writable is None:
            path = self.QUEUE_PREEMPTION_PARAM
            self._quantum_knob_writable = os.path.isfile(path) and os.access(
                path, os.W_OK
            )
            if self._quantum_knob_writable:
                try:
                    with open(path) as f:
                        self._quantum_default_ms = int(f.read().strip())
                except (OSError, ValueError):
                    self._quantum_knob_writable = False
        return self._quantum_knob_writable

    def timeslice_effective(self) -> bool:
        return self._probe_quantum_knob()

    def set_timeslice_quantum(self, gpu_index: int, quantum_us: Optional[int]) -> None:
        """Scheduler-quantum control (the compute-policy analog, SURVEY.md
        §2.3 N7). When the node exposes a writable
        ``queue_preemption_timeout_ms`` the quantum is applied there (a
        node-global knob: amdgpu has no per-GPU granularity, documented via
        the ``timeSlicingEffective`` attribute contract); otherwise the
        request is recorded and surfaced as advisory — HSA default
        time-slicing still multiplexes queues fairly."""
        others = {
            i: q
            for i, q in self._timeslice.items()
            if i != gpu_index and q is not None and q != quantum_us
        }
        self._timeslice[gpu_index] = quantum_us
        if self._probe_quantum_knob():
            if quantum_us is not None and others:
                # the knob is node-global: a different quantum is active
                # for another GPU's claim and this write overrides it
                log.warning(
                    "queue preemption timeout is node-global: setting "
                    "%s us for gpu-%d overrides active quanta %s",
                    quantum_us,
                    gpu_index,
                    others,
                )
            if quantum_us is None:
                # restore: fall back to another GPU's still-active
                # quantum (node-global knob), else the boot default
                active = sorted(
                    q for q in self._timeslice.values() if q is not None
                )
                value_ms = (
                    max(1, active[0] // 1000)
                    if active
                    else self._quantum_default_ms
                )
            else:
                value_ms = max(1, quantum_us // 1000)
            try:
                with open(self.QUEUE_PREEMPTION_PARAM, "w") as f:
                    f.write(str(value_ms))
                log.info(
                    "gpu-%d: queue preemption timeout set to %d ms "
                    "(node-global amdgpu knob)",
                    gpu_index,
                    value_ms,
                )
                return
            except OSError as e:
                log.warning("quantum knob write failed: %s", e)
                self._quantum_knob_writable = False
        log.info(
            "gpu-%d: time-slice quantum request %s recorded "
            "(amdgpu scheduler default multiplexing applies)",
            gpu_index,
            quantum_us,
        )

    def get_timeslice_quantum(self, gpu_index: int) -> Optional[int]:
        return self._timeslice.get(gpu_index)

    # -- device nodes --------------------------------------------------------
    def device_node_paths(
        self, gpu_index: int, partition_id: Optional[int] = None
    ) -> Dict[str, str]:
        gpus = self.enumerate()
        if gpu_index >= len(gpus):
            raise HalError(f"gpu-{gpu_index} not found")
        g = gpus[gpu_index]
        render, card = g.render_minor, g.card_minor
        if partition_id is not None and g.partitions:
            part = next(
                (p for p in g.partitions if p.partition_id == partition_id),
                None,
            )
            if part is None:
                raise HalError(
                    f"gpu-{gpu_index} has no partition {partition_id}"
                )
            render, card = part.render_minor, part.card_minor
        return {
            "kfd": "/dev/kfd",
            "renderD": f"/dev/dri/renderD{render}",
            "card": f"/dev/dri/card{card}" if card >= 0 else "",
        }

    # -- health --------------------------------------------------------------
    def health_check(self, gpu_index: int) -> Dict[str, str]:
        gpus = self.enumerate()
        if gpu_index >= len(gpus):
            return {"status": "missing"}
        g = gpus[gpu_index]
        return {
            "status": "healthy",
            "uuid": g.uuid,
            "computePartition": g.compute_partition,
            "memoryPartition": g.memory_partition,
            "renderD": str(g.render_minor),
        }
