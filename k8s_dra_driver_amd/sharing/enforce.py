"""Shared-GPU isolation enforcement.

``HSA_CU_MASK`` / ``GPU_MAX_ALLOC_PERCENT`` are environment variables a
workload can unset; the reference's MPS daemon enforces its limits
out-of-band (``sharing.go:211-221``). amdgpu has no daemon to do that, but
the node plugin runs as root and KFD exposes every GPU-attached process
under ``/sys/class/kfd/kfd/proc/<pid>/``, so the supervisor can close the
loop the other way: **detect** any process that belongs to a SharedCompute
claim (attributed via the CDI-injected ``AMD_DRA_SHARED_SESSION`` in
``/proc/<pid>/environ``) whose effective environment no longer carries the
session's CU mask — i.e. a container that stripped or altered its slice —
and act on it (Warning event / metric / SIGKILL in ``kill`` mode).

Design notes:
- /proc/<pid>/environ shows the *initial* env; a process can mutate its
  own copy afterwards, but ROCR reads HSA_CU_MASK once at runtime init,
  so the initial env is exactly what the HSA runtime honored. A workload
  that execs itself with a scrubbed env re-materializes in /proc with the
  scrubbed environ and is caught here.
- Processes with GPU access but no session attribution ("foreign": e.g.
  host processes outside any claim) are reported distinctly; policy for
  them belongs to the operator.
"""

from __future__ import annotations

import logging
import os
import signal
import threading
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

from .shared import SharedComputeManager

log = logging.getLogger(__name__)

KFD_PROC_ROOT = "/sys/class/kfd/kfd/proc"


@dataclass
class Violation:
    pid: int
    kind: str  # "stripped" | "altered" | "orphan-session"
    session_id: str
    claim_uid: str
    detail: str


def _read_environ(proc_root: str, pid: int) -> Optional[Dict[str, str]]:
    try:
        with open(f"{proc_root}/{pid}/environ", "rb") as f:
            raw = f.read()
    except OSError:
        return None  # exited or not ours to read
    env: Dict[str, str] = {}
    for entry in raw.split(b"\0"):
        if b"=" in entry:
            k, _, v = entry.partition(b"=")
            env[k.decode(errors="replace")] = v.decode(errors="replace")
    return env


class SharedEnforcer:
    """Periodic scanner binding KFD's process list to SharedCompute
    sessions; the out-of-band enforcement loop MPS gets from its daemon."""

    def __init__(
        self,
        manager: SharedComputeManager,
        *,
        proc_root: str = "/proc",
        kfd_proc_root: str = KFD_PROC_ROOT,
        action: str = "warn",  # "warn" | "kill"
        interval_s: float = 10.0,
        on_violation: Optional[Callable[[Violation], None]] = None,
    ):
        if action not in ("warn", "kill"):
            raise ValueError(f"action must be warn|kill, got {action!r}")
        self.manager = manager
        self.proc_root = proc_root
        self.kfd_proc_root = kfd_proc_root
        self.action = action
        self.interval_s = interval_s
        self.on_violation = on_violation
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        #: observability: total violations seen (metrics surface)
        self.violation_count = 0
        #: KFD pids with no /proc entry in the last scan — non-empty means
        #: this process lacks the host PID namespace (deploy with
        #: hostPID: true, as the Helm chart does) and cannot attribute
        self.last_unattributable: List[int] = []
        self._warned_ns = False

    # -- scanning ----------------------------------------------------------
    def gpu_pids(self) -> List[int]:
        try:
            entries = os.listdir(self.kfd_proc_root)
        except OSError:
            return []  # no KFD (CPU node) — nothing to enforce
        return sorted(int(e) for e in entries if e.isdigit())

    def _expected_masks(self, session) -> List[str]:
        return sorted(
            e.split("=", 1)[1]
            for e in session.env
            if e.startswith("HSA_CU_MASK=")
        )

    def scan(self) -> List[Violation]:
        """One pass over GPU-attached processes; returns violations found
        (also dispatched to ``on_violation`` / acted on per ``action``)."""
        out: List[Violation] = []
        unattributable: List[int] = []
        for pid in self.gpu_pids():
            env = _read_environ(self.proc_root, pid)
            if env is None:
                if not os.path.isdir(f"{self.proc_root}/{pid}"):
                    unattributable.append(pid)
                continue
            session_id = env.get("AMD_DRA_SHARED_SESSION", "")
            if not session_id:
                continue  # not a SharedCompute claim process
            claim_uid = env.get("AMD_DRA_CLAIM_UID", "")
            session = self.manager.get_session(session_id)
            if session is None:
                out.append(
                    Violation(
                        pid,
                        "orphan-session",
                        session_id,
                        claim_uid,
                        "process claims a session the supervisor does not "
                        "know (leak across plugin restart or forged env)",
                    )
                )
                continue
            expected = self._expected_masks(session)
            if not expected:
                continue  # session without CU share: nothing to enforce
            actual = env.get("HSA_CU_MASK")
            if actual is None:
                out.append(
                    Violation(
                        pid,
                        "stripped",
                        session_id,
                        claim_uid,
                        f"HSA_CU_MASK stripped (expected one of {expected})",
                    )
                )
            elif actual not in expected:
                out.append(
                    Violation(
                        pid,
                        "altered",
                        session_id,
                        claim_uid,
                        f"HSA_CU_MASK={actual!r} not the assigned slice "
                        f"(expected one of {expected})",
                    )
                )
        self.last_unattributable = unattributable
        if unattributable and not self._warned_ns:
            self._warned_ns = True
            log.warning(
                "%d GPU-attached KFD pid(s) have no /proc entry — this "
                "process is likely missing the host PID namespace, so "
                "shared-GPU isolation cannot be attributed (deploy the "
                "plugin with hostPID: true; the Helm chart sets it)",
                len(unattributable),
            )
        for v in out:
            self._handle(v)
        return out

    def _handle(self, v: Violation) -> None:
        self.violation_count += 1
        log.warning(
            "shared-GPU isolation violation: pid=%d kind=%s session=%s "
            "claim=%s: %s",
            v.pid,
            v.kind,
            v.session_id,
            v.claim_uid,
            v.detail,
        )
        if self.on_violation is not None:
            try:
                self.on_violation(v)
            except Exception:
                log.exception("violation callback failed")
        if self.action == "kill" and v.kind in ("stripped", "altered"):
            try:
                os.kill(v.pid, signal.SIGKILL)
                log.warning("killed pid %d (enforcement action)", v.pid)
            except OSError as e:
                log.warning("kill %d failed: %s", v.pid, e)

    # -- lifecycle ---------------------------------------------------------
    def start(self) -> None:
        if self._thread is not None:
            return
        self._stop.clear()
        self._thread = threading.Thread(
            target=self._loop, name="shared-enforcer", daemon=True
        )
        self._thread.start()

    def _loop(self) -> None:
        while not self._stop.wait(self.interval_s):
            try:
                self.scan()
            except Exception:
                log.exception("enforcement scan failed")

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
