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

import logging
import threading
from typing import Callable, Dict, List, Optional

from ..hal.base import DeviceLib, HalError
from .catalog import DEFAULT_VALID_NPS, validate_mode_combo

log = logging.getLogger(__name__)


class RepartitionFailed(HalError):
    """A mode switch failed mid-sequence. ``original`` is the mode the GPU
    SHOULD return to; ``reverted`` says whether the best-effort revert
    already restored it (False = the GPU is in an intermediate mode and
    the caller must schedule a deferred restore)."""

    def __init__(self, msg: str, original: tuple, reverted: bool):
        super().__init__(msg)
        self.original = original
        self.reverted = reverted


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
            try:
                self._switch(gpu_index, cur_c, cur_m, compute_mode, memory_mode)
            except Exception as e:
                # best-effort revert to the original mode (the switch is
                # not atomic: a failure after the compute change leaves
                # an intermediate state)
                reverted = False
                try:
                    now = {g.index: g for g in self.lib.enumerate()}[gpu_index]
                    actual = (now.compute_partition, now.memory_partition)
                    if actual == (cur_c, cur_m):
                        reverted = True
                    else:
                        self._switch(
                            gpu_index, actual[0], actual[1], cur_c, cur_m
                        )
                        reverted = True
                except Exception:
                    log.exception(
                        "gpu-%d: revert after failed switch also failed; "
                        "left in an intermediate mode (deferred restore "
                        "required)",
                        gpu_index,
                    )
                raise RepartitionFailed(
                    f"gpu-{gpu_index} mode switch to "
                    f"{compute_mode}/{memory_mode} failed: {e}",
                    original=(cur_c, cur_m),
                    reverted=reverted,
                ) from e
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
