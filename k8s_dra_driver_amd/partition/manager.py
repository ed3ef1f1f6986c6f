"""Dynamic partition reconfiguration engine.

The capability the reference could not ship: its dynamic MIG create/delete
is committed but commented out pending Kubernetes support
(``nvlib.go:560-669``, ``device_state.go:512-558``). On MI355X, partition
switches ARE dynamic (``amdsmi_set_gpu_compute_partition`` /
``..._memory_partition``) — but they affect the *whole GPU* and change the
visible KFD/DRM device set, so doing one inside Prepare needs an ownership
model (SURVEY.md §7 hard-part 1):

- per-GPU mutex: at most one reconfiguration in flight per GPU;
- drain guard: a GPU is reconfigurable only when no *other* prepared claim
  holds any device backed by it (an in-use oracle injected by DeviceState);
- mode-change ordering: NPS and compute modes constrain each other, so the
  engine sequences through a compatible intermediate (NPS1) when needed;
- post-switch re-enumeration is the caller's job (DeviceState refreshes its
  allocatable map and republishes ResourceSlices via the HAL observer).
"""

from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

from ..hal.base import DeviceLib, HalError
from .catalog import DEFAULT_VALID_NPS, validate_mode_combo


class RepartitionRefused(HalError):
    """Raised when a repartition is structurally or operationally invalid."""


class PartitionManager:
    def __init__(
        self,
        lib: DeviceLib,
        *,
        in_use_fn: Optional[Callable[[int], List[str]]] = None,
    ):
        """``in_use_fn(gpu_index)`` -> claim UIDs (other than the current
        one) holding devices backed by that GPU."""
        self.lib = lib
        self.in_use_fn = in_use_fn or (lambda idx: [])
        self._gpu_locks: Dict[int, threading.Lock] = {}
        self._registry_lock = threading.Lock()

    def _lock_for(self, gpu_index: int) -> threading.Lock:
        with self._registry_lock:
            return self._gpu_locks.setdefault(gpu_index, threading.Lock())

    def ensure_mode(
        self,
        gpu_index: int,
        compute_mode: str,
        memory_mode: str,
        *,
        requesting_claim: str = "",
        allow_dynamic: bool = False,
    ) -> bool:
        """Bring the GPU to (compute_mode, memory_mode).

        Returns True if a switch happened, False if already in the mode.
        Raises RepartitionRefused if a switch is needed but not allowed or
        the GPU is held by other claims.
        """
        validate_mode_combo(compute_mode, memory_mode)
        lock = self._lock_for(gpu_index)
        with lock:
            gpus = {g.index: g for g in self.lib.enumerate()}
            if gpu_index not in gpus:
                raise RepartitionRefused(f"gpu-{gpu_index} not found")
            gpu = gpus[gpu_index]
            cur_c, cur_m = gpu.compute_partition, gpu.memory_partition
            if (cur_c, cur_m) == (compute_mode, memory_mode):
                return False
            if not allow_dynamic:
                raise RepartitionRefused(
                    f"gpu-{gpu_index} is {cur_c}/{cur_m} but the claim needs "
                    f"{compute_mode}/{memory_mode} and dynamic repartition "
                    f"was not allowed (set allowDynamicRepartition)"
                )
            holders = [
                uid for uid in self.in_use_fn(gpu_index) if uid != requesting_claim
            ]
            if holders:
                raise RepartitionRefused(
                    f"gpu-{gpu_index} repartition refused: held by "
                    f"{len(holders)} other prepared claim(s) {holders[:3]}"
                )
            self._switch(gpu_index, cur_c, cur_m, compute_mode, memory_mode)
            return True

    def _switch(
        self,
        gpu_index: int,
        cur_c: str,
        cur_m: str,
        new_c: str,
        new_m: str,
    ) -> None:
        """Sequence the mode changes through compatible intermediates.

        Invariant: after every individual HAL call the (compute, NPS) pair
        is valid. NPS1 is valid under every compute mode, so it is the
        universal bridge.
        """
        if cur_c != new_c:
            if cur_m not in DEFAULT_VALID_NPS[new_c]:
                # bridge through NPS1 before the compute switch
                self.lib.set_memory_partition(gpu_index, "NPS1")
                cur_m = "NPS1"
            self.lib.set_compute_partition(gpu_index, new_c)
            cur_c = new_c
        if cur_m != new_m:
            self.lib.set_memory_partition(gpu_index, new_m)
