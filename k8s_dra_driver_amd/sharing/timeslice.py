"""Time-slicing manager (reference TimeSlicingManager, sharing.go:97-122).

Maps the config-surface intervals (Default/Short/Medium/Long) onto the
amdgpu scheduler quantum through the HAL. Mirrors the reference's rules:
time-slicing applies to whole GPUs only (a compute partition shares its
parent's scheduler), and unprepare restores the default interval
(``device_state.go:358-362``).

Where the reference shells out to ``nvidia-smi compute-policy`` once per GPU
sequentially (``nvlib.go:521-539``, part of its prepare-latency problem),
this manager issues one in-process HAL call per distinct parent GPU.
"""

from __future__ import annotations

from typing import Iterable, List

from ..api.types import TimeSlicingSettings
from ..hal.base import DeviceLib
from ..hal.model import AllocatableDevice


class SharingError(RuntimeError):
    pass


class TimeSlicingManager:
    def __init__(self, lib: DeviceLib):
        self.lib = lib

    def set_timeslice(
        self, devices: Iterable[AllocatableDevice], settings: TimeSlicingSettings
    ) -> List[int]:
        """Apply the interval to every distinct parent GPU; returns the GPU
        indices touched (recorded in the checkpoint for unprepare)."""
        settings.validate()
        gpu_indices: List[int] = []
        for dev in devices:
            if dev.kind != "gpu":
                raise SharingError(
                    f"time-slicing is supported on whole GPUs only; "
                    f"{dev.canonical_name} is a compute partition "
                    f"(its parent's scheduler is shared)"
                )
            idx = dev.parent_gpu.index
            if idx not in gpu_indices:
                gpu_indices.append(idx)
        for idx in gpu_indices:
            self.lib.set_timeslice_quantum(idx, settings.quantum_us)
        return gpu_indices

    def restore_default(self, gpu_indices: Iterable[int]) -> None:
        for idx in gpu_indices:
            self.lib.set_timeslice_quantum(idx, None)
