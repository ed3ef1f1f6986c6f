"""Shared-GPU supervisor — the MPS-control-daemon analog, node-local.

The reference implements spatial sharing by creating a per-claim
**Deployment** running ``nvidia-cuda-mps-control``, mounting a tmpfs shm
sized MemTotal/2, flipping the GPU to EXCLUSIVE_PROCESS, then polling the
Deployment for readiness with exponential backoff — seconds of latency on
the prepare hot path (``sharing.go:124-403``).

AMD has no MPS daemon: concurrent HSA queues from multiple processes are the
default. What a shared-GPU claim needs is *scoping*, which this supervisor
provides in-process with zero daemon or API-server round-trips
(SURVEY.md §7 step 6 "avoid the Deployment+poll design"):

- a per-claim shm/IPC directory bind-mounted into every container of the
  claim (tmpfs when running as root on a real node, plain dir otherwise),
- **CU masking** via ``HSA_CU_MASK`` — the supervisor carves disjoint CU
  ranges per claim sharing one GPU, giving each claim a guaranteed slice of
  the 256 CUs (the active-thread-percentage analog),
- best-effort VRAM budgeting env (``GPU_MAX_ALLOC_PERCENT`` + an
  introspection variable per device); hard VRAM isolation has no amdgpu
  primitive today (SURVEY.md §7 hard-part 4) and is documented best-effort.
"""

from __future__ import annotations

import os
import shutil
import subprocess
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..api.types import SharedComputeSettings
from ..cdi.spec import ContainerEdits, Mount
from ..hal.model import AllocatableDevice

DEFAULT_SUPERVISOR_ROOT = "/run/amd-dra/shared"


@dataclass
class _GpuShareState:
    """CU-range bookkeeping for one GPU: session_id -> (start_cu, num_cu)."""

    cu_count: int = 256
    ranges: Dict[str, tuple] = field(default_factory=dict)

    def allocate(self, session_id: str, num_cu: int) -> tuple:
        """First-fit allocation of a contiguous CU range."""
        used = sorted(self.ranges.values())
        start = 0
        for s, n in used:
            if start + num_cu <= s:
                break
            start = s + n
        if start + num_cu > self.cu_count:
            raise RuntimeError(
                f"no CU capacity left: want {num_cu}, "
                f"used {sum(n for _, n in used)}/{self.cu_count}"
            )
        self.ranges[session_id] = (start, num_cu)
        return (start, num_cu)

    def release(self, session_id: str) -> None:
        self.ranges.pop(session_id, None)


def cu_mask_hex(start: int, count: int, total: int = 256) -> str:
    """HSA_CU_MASK hex string selecting CUs [start, start+count)."""
    mask = ((1 << count) - 1) << start
    width = (total + 3) // 4
    return f"0x{mask:0{width}x}"


@dataclass
class SharedSession:
    session_id: str
    claim_uid: str
    shm_dir: str
    env: List[str]
    mounts: List[Mount]
    gpu_indices: List[int]

    def container_edits(self) -> ContainerEdits:
        return ContainerEdits(env=list(self.env), mounts=list(self.mounts))


class SharedComputeManager:
    """Node-local supervisor owning shared-GPU claim sessions."""

    def __init__(
        self,
        root: str = DEFAULT_SUPERVISOR_ROOT,
        *,
        use_tmpfs: Optional[bool] = None,
        shm_size_mb: int = 65536,
    ):
        self.root = root
        self.shm_size_mb = shm_size_mb
        # tmpfs requires root + mount(2); auto-detect, tests force False.
        self.use_tmpfs = use_tmpfs if use_tmpfs is not None else (os.geteuid() == 0)
        self._lock = threading.Lock()
        self._sessions: Dict[str, SharedSession] = {}
        self._gpu_state: Dict[int, _GpuShareState] = {}

    # -- session lifecycle -------------------------------------------------
    def start_session(
        self,
        claim_uid: str,
        devices: List[AllocatableDevice],
        settings: SharedComputeSettings,
    ) -> SharedSession:
        """Create the per-claim shared scope. Synchronous and local: no
        daemon spawn, no readiness poll (vs sharing.go:289-344)."""
        settings.validate()
        session_id = claim_uid[:36]
        with self._lock:
            if session_id in self._sessions:
                return self._sessions[session_id]

            shm_dir = os.path.join(self.root, session_id, "shm")
            os.makedirs(shm_dir, exist_ok=True)
            if self.use_tmpfs:
                self._mount_tmpfs(shm_dir)

            env = ["AMD_DRA_SHARED=1", f"AMD_DRA_SHARED_SESSION={session_id}"]
            mounts = [
                Mount(
                    host_path=shm_dir,
                    container_path="/dev/shm",
                    options=["rw", "nosuid", "nodev", "bind"],
                )
            ]
            gpu_indices: List[int] = []
            allocated: List[int] = []
            cu_entries: List[str] = []
            try:
                for dev in devices:
                    gpu = dev.parent_gpu
                    if gpu.index in gpu_indices:
                        continue
                    gpu_indices.append(gpu.index)
                    share = settings.default_cu_share_percent
                    if share is not None:
                        st = self._gpu_state.setdefault(
                            gpu.index, _GpuShareState(cu_count=gpu.cu_count)
                        )
                        num_cu = max(1, gpu.cu_count * share // 100)
                        start, count = st.allocate(session_id, num_cu)
                        allocated.append(gpu.index)
                        cu_entries.append(
                            f"{gpu.index}:"
                            f"{cu_mask_hex(start, count, gpu.cu_count)}"
                        )
                # ONE env var, ';'-joined per GPU: multiple same-name env
                # entries collapse in a real container environment, and the
                # joined form is hardware-verified to constrain each listed
                # GPU (profiles/cu_mask_syntax_r02: hex single, hex joined
                # and decimal ranges all quarter MFMA throughput on gfx950)
                if cu_entries:
                    env.append("HSA_CU_MASK=" + ";".join(cu_entries))
                # VRAM budgeting (best-effort; see module docstring)
                uuids_by_index = {
                    i: d.uuid for i, d in enumerate(devices)
                }
                limits = settings.normalized_memory_limits(uuids_by_index)
                for uuid, limit in sorted(limits.items()):
                    env.append(f"AMD_DRA_MEMORY_LIMIT_{uuid.replace('-', '_')}={limit}")
                if settings.default_memory_limit is not None and devices:
                    vram = devices[0].parent_gpu.vram_total_mib * 1024 * 1024
                    pct = min(
                        100,
                        max(
                            1,
                            100
                            * (limits.get(devices[0].uuid) or vram)
                            // vram,
                        ),
                    )
                    env.append(f"GPU_MAX_ALLOC_PERCENT={pct}")
            except BaseException:
                for idx in allocated:
                    self._gpu_state[idx].release(session_id)
                shutil.rmtree(os.path.join(self.root, session_id), ignore_errors=True)
                raise

            session = SharedSession(
                session_id=session_id,
                claim_uid=claim_uid,
                shm_dir=shm_dir,
                env=env,
                mounts=mounts,
                gpu_indices=gpu_indices,
            )
            self._sessions[session_id] = session
            return session

    def stop_session(self, session_id: str) -> None:
        """Teardown (sharing.go:368-403 analog); idempotent."""
        with self._lock:
            session = self._sessions.pop(session_id, None)
            for st in self._gpu_state.values():
                st.release(session_id)
            base = os.path.join(self.root, session_id)
            if session and self.use_tmpfs:
                self._umount_tmpfs(session.shm_dir)
            shutil.rmtree(base, ignore_errors=True)

    def recover_session(self, session: SharedSession) -> None:
        """Re-register a session from a checkpoint after plugin restart."""
        with self._lock:
            self._sessions[session.session_id] = session
            for e in session.env:
                if e.startswith("HSA_CU_MASK="):
                    # parse "<idx>:0x<mask>[;<idx>:0x<mask>...]" back into
                    # range bookkeeping (one ';'-joined var per session)
                    for entry in e.split("=", 1)[1].split(";"):
                        if ":" not in entry:
                            continue
                        idx_s, mask_s = entry.split(":", 1)
                        mask = int(mask_s, 16)
                        start = (mask & -mask).bit_length() - 1 if mask else 0
                        count = bin(mask).count("1")
                        st = self._gpu_state.setdefault(
                            int(idx_s), _GpuShareState()
                        )
                        st.ranges[session.session_id] = (start, count)

    def get_session(self, session_id: str) -> Optional[SharedSession]:
        with self._lock:
            return self._sessions.get(session_id)

    # -- tmpfs helpers -----------------------------------------------------
    def _mount_tmpfs(self, path: str) -> None:
        subprocess.run(
            [
                "mount",
                "-t",
                "tmpfs",
                "-o",
                f"rw,nosuid,nodev,size={self.shm_size_mb}m",
                "tmpfs",
                path,
            ],
            check=True,
            capture_output=True,
        )

    def _umount_tmpfs(self, path: str) -> None:
        subprocess.run(["umount", path], check=False, capture_output=True)
