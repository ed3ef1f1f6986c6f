"""GPU health monitoring: failure detection + ResourceSlice self-healing.

The reference has no device health path at all — a dead GPU stays
advertised until the plugin restarts (SURVEY.md §5.3 lists only claim-level
retry loops). This monitor:

- polls ``DeviceLib.health_check`` per GPU on an interval,
- on failure, excludes that GPU's devices from publication and republishes
  (scheduler stops placing new claims there),
- on recovery, re-includes them,
- exposes state for metrics and node conditions.

The fake HAL's fault injector doubles as the test harness.
"""

from __future__ import annotations

import logging
import threading
from typing import Callable, Dict, Optional, Set

from ..hal.base import DeviceLib

log = logging.getLogger(__name__)


class HealthMonitor:
    def __init__(
        self,
        lib: DeviceLib,
        *,
        on_change: Optional[Callable[[Set[int]], None]] = None,
        interval_s: float = 30.0,
        failures_to_unhealthy: int = 2,
        successes_to_healthy: int = 1,
    ):
        self.lib = lib
        self.on_change = on_change
        self.interval_s = interval_s
        self.failures_to_unhealthy = failures_to_unhealthy
        self.successes_to_healthy = successes_to_healthy
        self._fail_counts: Dict[int, int] = {}
        self._ok_counts: Dict[int, int] = {}
        self._unhealthy: Set[int] = set()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    @property
    def unhealthy_gpus(self) -> Set[int]:
        with self._lock:
            return set(self._unhealthy)

    def check_once(self) -> Set[int]:
        """One poll over all GPUs; returns the unhealthy set after
        hysteresis. Fires on_change when the set changes."""
        try:
            gpus = self.lib.enumerate()
        except Exception:
            log.exception("health enumeration failed")
            return self.unhealthy_gpus
        changed = False
        with self._lock:
            for g in gpus:
                ok = False
                try:
                    ok = self.lib.health_check(g.index).get("status") == "healthy"
                except Exception as e:
                    log.warning("gpu-%d health check error: %s", g.index, e)
                if ok:
                    self._ok_counts[g.index] = self._ok_counts.get(g.index, 0) + 1
                    self._fail_counts[g.index] = 0
                    if (
                        g.index in self._unhealthy
                        and self._ok_counts[g.index] >= self.successes_to_healthy
                    ):
                        self._unhealthy.discard(g.index)
                        changed = True
                        log.info("gpu-%d recovered; republishing", g.index)
                else:
                    self._fail_counts[g.index] = (
                        self._fail_counts.get(g.index, 0) + 1
                    )
                    self._ok_counts[g.index] = 0
                    if (
                        g.index not in self._unhealthy
                        and self._fail_counts[g.index]
                        >= self.failures_to_unhealthy
                    ):
                        self._unhealthy.add(g.index)
                        changed = True
                        log.error(
                            "gpu-%d marked unhealthy after %d failures",
                            g.index,
                            self._fail_counts[g.index],
                        )
            result = set(self._unhealthy)
        if changed and self.on_change is not None:
            self.on_change(result)
        return result

    def run(self) -> None:
        while not self._stop.is_set():
            self.check_once()
            self._stop.wait(self.interval_s)

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self.run, name="gpu-health", daemon=True
        )
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
