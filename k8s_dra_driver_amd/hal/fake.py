"""Fake 8xMI355X hardware backend with partition state machine + faults.

The reference has no hardware-free test backend at all — its
enumeration/prepare logic is only exercisable on real NVIDIA nodes. This
backend models one 8xMI355X OAM node faithfully enough that the entire driver
(enumeration, ResourceSlice publication, prepare/unprepare, dynamic
repartition, topology-aware allocation) runs and is testable with zero GPUs:

- 8 GPUs, OAM 0-7, fully connected xGMI mesh (7 p2p links per GPU),
- per-GPU partition state machine: SPX/DPX/QPX/CPX x NPS1/NPS4, with render
  nodes appearing/disappearing on mode switches exactly like KFD does,
- fault injection knobs per operation (error or latency), which doubles as
  the failure-detection test harness.
"""

from __future__ import annotations

import copy
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..partition.catalog import (
    COMPUTE_MODES,
    DEFAULT_VALID_NPS,
    GFX950_CU_COUNT,
    GFX950_VRAM_MIB,
    GFX950_XCD_COUNT,
    make_profile,
    validate_mode_combo,
)
from .base import DeviceLib, HalError, HalNotSupported
from .model import GpuInfo, PartitionedDeviceInfo, XgmiLink


@dataclass
class FaultInjector:
    """Per-operation fault/latency injection (SURVEY.md §5.3 'Build').

    ``errors[op]`` -> exception raised on the next call(s) of ``op``;
    ``latency_s[op]`` -> sleep injected into every call of ``op``.
    ``op`` is the DeviceLib method name.
    """

    errors: Dict[str, List[Exception]] = field(default_factory=dict)
    latency_s: Dict[str, float] = field(default_factory=dict)
    call_counts: Dict[str, int] = field(default_factory=dict)

    def fail_next(self, op: str, exc: Exception, times: int = 1) -> None:
        self.errors.setdefault(op, []).extend([exc] * times)

    def set_latency(self, op: str, seconds: float) -> None:
        self.latency_s[op] = seconds

    def fire(self, op: str) -> None:
        self.call_counts[op] = self.call_counts.get(op, 0) + 1
        delay = self.latency_s.get(op, 0.0)
        if delay:
            time.sleep(delay)
        queue = self.errors.get(op)
        if queue:
            raise queue.pop(0)


@dataclass
class FakeNodeConfig:
    """Shape of the modeled node (defaults: one 8xMI355X OAM baseboard)."""

    num_gpus: int = 8
    vram_total_mib: int = GFX950_VRAM_MIB
    cu_count: int = GFX950_CU_COUNT
    xcd_count: int = GFX950_XCD_COUNT
    driver_version: str = "6.14.5"
    rocm_version: str = "7.2.0"
    hive_id: str = "hive-0x5a1ad"
    #: render minor of gpu 0 in SPX; kfd render nodes start at 128 on amdgpu
    base_render_minor: int = 128


class _FakeGpuState:
    """Mutable partition state of one fake GPU."""

    def __init__(self, cfg: FakeNodeConfig, index: int):
        self.cfg = cfg
        self.index = index
        self.compute_mode = "SPX"
        self.memory_mode = "NPS1"
        self.uuid = f"amd-mi355x-{index:02d}-{0xACE0 + index:04x}"
        self.oam_id = index
        self.busy_pids: List[int] = []  # processes holding the GPU


class FakeDeviceLib(DeviceLib):
    """In-memory DeviceLib over a modeled 8xMI355X node."""

    def __init__(
        self,
        config: Optional[FakeNodeConfig] = None,
        faults: Optional[FaultInjector] = None,
    ):
        self.cfg = config or FakeNodeConfig()
        self.faults = faults or FaultInjector()
        self._lock = threading.RLock()
        self._open = False
        self._gpus = [_FakeGpuState(self.cfg, i) for i in range(self.cfg.num_gpus)]
        self._timeslice: Dict[int, Optional[int]] = {}
        #: models whether this fake node honors quantum requests (set False
        #: to simulate real amdgpu pools where time-slicing is advisory)
        self.timeslice_effective_flag = True
        #: models bare-metal partition-switch capability
        self.repartition_capable_flag = True
        #: observers notified after any partition-state change (used by the
        #: plugin to republish ResourceSlices without polling)
        self._observers: List[Callable[[], None]] = []

    # -- lifecycle ---------------------------------------------------------
    def open(self) -> None:
        self.faults.fire("open")
        self._open = True

    def close(self) -> None:
        self._open = False

    def _check_open(self) -> None:
        if not self._open:
            raise HalError("device library not open")

    def subscribe(self, fn: Callable[[], None]) -> None:
        self._observers.append(fn)

    def _notify(self) -> None:
        for fn in list(self._observers):
            fn()

    # -- test helpers ------------------------------------------------------
    def mark_busy(self, gpu_index: int, pid: int) -> None:
        """Simulate a process holding the GPU (blocks repartition)."""
        with self._lock:
            self._gpus[gpu_index].busy_pids.append(pid)

    def mark_idle(self, gpu_index: int, pid: Optional[int] = None) -> None:
        with self._lock:
            g = self._gpus[gpu_index]
            if pid is None:
                g.busy_pids.clear()
            elif pid in g.busy_pids:
                g.busy_pids.remove(pid)

    # -- enumeration -------------------------------------------------------
    def _render_minor(self, gpu_index: int, partition_id: int) -> int:
        """Deterministic render-minor layout mirroring KFD behavior:
        each GPU owns a contiguous block of 8 minors; partition p of GPU g
        sits at base + g*8 + p (SPX uses slot 0)."""
        return self.cfg.base_render_minor + gpu_index * 8 + partition_id

    def _kfd_node(self, gpu_index: int, partition_id: int) -> int:
        return gpu_index * 8 + partition_id + 1  # node 0 is the CPU

    def enumerate(self) -> List[GpuInfo]:
        self.faults.fire("enumerate")
        self._check_open()
        with self._lock:
            out: List[GpuInfo] = []
            for g in self._gpus:
                links = [
                    XgmiLink(
                        peer_oam_id=p.oam_id,
                        peer_uuid=p.uuid,
                        num_lanes=16,
                        max_bandwidth_gbps=153,
                    )
                    for p in self._gpus
                    if p.index != g.index
                ]
                info = GpuInfo(
                    index=g.index,
                    uuid=g.uuid,
                    oam_id=g.oam_id,
                    pcie_bdf=f"0000:{0x0c + g.index * 0x10:02x}:00.0",
                    vram_total_mib=self.cfg.vram_total_mib,
                    cu_count=self.cfg.cu_count,
                    xcd_count=self.cfg.xcd_count,
                    driver_version=self.cfg.driver_version,
                    rocm_version=self.cfg.rocm_version,
                    kfd_node_id=self._kfd_node(g.index, 0),
                    render_minor=self._render_minor(g.index, 0),
                    card_minor=g.index,
                    compute_partition=g.compute_mode,
                    memory_partition=g.memory_mode,
                    nps_caps=list(DEFAULT_VALID_NPS[g.compute_mode]),
                    compute_caps=list(COMPUTE_MODES),
                    xgmi_hive_id=self.cfg.hive_id,
                    xgmi_node_id=g.index,
                    timeslice_effective=self.timeslice_effective_flag,
                    repartition_capable=self.repartition_capable_flag,
                    links=links,
                )
                if g.compute_mode != "SPX":
                    prof = make_profile(
                        g.compute_mode,
                        g.memory_mode,
                        vram_total_mib=self.cfg.vram_total_mib,
                        cu_count=self.cfg.cu_count,
                        xcd_count=self.cfg.xcd_count,
                        nps_caps=DEFAULT_VALID_NPS[g.compute_mode],
                    )
                    for pid in range(prof.num_partitions):
                        info.partitions.append(
                            PartitionedDeviceInfo(
                                parent_index=g.index,
                                parent_uuid=g.uuid,
                                partition_id=pid,
                                profile=prof,
                                kfd_node_id=self._kfd_node(g.index, pid),
                                render_minor=self._render_minor(g.index, pid),
                                card_minor=g.index,
                            )
                        )
                out.append(copy.deepcopy(info))
            return out

    # -- partitioning ------------------------------------------------------
    def set_compute_partition(self, gpu_index: int, mode: str) -> None:
        self.faults.fire("set_compute_partition")
        self._check_open()
        with self._lock:
            g = self._gpus[gpu_index]
            if g.busy_pids:
                raise HalError(
                    f"gpu-{gpu_index}: compute partition switch refused, "
                    f"{len(g.busy_pids)} process(es) still using the GPU"
                )
            # Validate against the *target* mode's NPS compatibility; if the
            # current NPS mode is invalid for the new compute mode, KFD
            # refuses — callers must set NPS first (manager handles ordering).
            validate_mode_combo(mode, "NPS1")  # mode name check
            if g.memory_mode not in DEFAULT_VALID_NPS[mode]:
                raise HalError(
                    f"gpu-{gpu_index}: {mode} invalid under current memory "
                    f"mode {g.memory_mode} (allowed {DEFAULT_VALID_NPS[mode]})"
                )
            if mode != g.compute_mode:
                g.compute_mode = mode
                self._notify()

    def set_memory_partition(self, gpu_index: int, mode: str) -> None:
        self.faults.fire("set_memory_partition")
        self._check_open()
        with self._lock:
            g = self._gpus[gpu_index]
            if g.busy_pids:
                raise HalError(
                    f"gpu-{gpu_index}: memory partition switch refused, "
                    f"GPU busy"
                )
            try:
                validate_mode_combo(g.compute_mode, mode, nps_caps=("NPS1", "NPS2", "NPS4"))
            except ValueError as e:
                raise HalError(f"gpu-{gpu_index}: {e}") from e
            if mode not in DEFAULT_VALID_NPS[g.compute_mode]:
                raise HalError(
                    f"gpu-{gpu_index}: {mode} invalid under compute mode "
                    f"{g.compute_mode}"
                )
            if mode != g.memory_mode:
                g.memory_mode = mode
                self._notify()

    # -- scheduler controls --------------------------------------------------
    def set_timeslice_quantum(self, gpu_index: int, quantum_us: Optional[int]) -> None:
        self.faults.fire("set_timeslice_quantum")
        self._check_open()
        if quantum_us is not None and quantum_us <= 0:
            raise HalNotSupported(f"invalid quantum {quantum_us}")
        self._timeslice[gpu_index] = quantum_us

    def get_timeslice_quantum(self, gpu_index: int) -> Optional[int]:
        return self._timeslice.get(gpu_index)

    def timeslice_effective(self) -> bool:
        return self.timeslice_effective_flag

    def dynamic_repartition_capable(self) -> bool:
        return self.repartition_capable_flag

    # -- device nodes --------------------------------------------------------
    def device_node_paths(
        self, gpu_index: int, partition_id: Optional[int] = None
    ) -> Dict[str, str]:
        self.faults.fire("device_node_paths")
        self._check_open()
        with self._lock:
            g = self._gpus[gpu_index]
            pid = partition_id or 0
            if partition_id is not None and g.compute_mode == "SPX" and partition_id != 0:
                raise HalError(
                    f"gpu-{gpu_index} is SPX; partition {partition_id} does not exist"
                )
            return {
                "kfd": "/dev/kfd",
                "renderD": f"/dev/dri/renderD{self._render_minor(gpu_index, pid)}",
                "card": f"/dev/dri/card{g.index}",
            }

    # -- health --------------------------------------------------------------
    def health_check(self, gpu_index: int) -> Dict[str, str]:
        self.faults.fire("health_check")
        self._check_open()
        g = self._gpus[gpu_index]
        return {
            "status": "healthy",
            "uuid": g.uuid,
            "computePartition": g.compute_mode,
            "memoryPartition": g.memory_mode,
        }
