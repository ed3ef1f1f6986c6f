"""Crash-safe prepared-claim checkpoints — per-claim files, not a full map.

The reference persists ONE checksummed JSON map of all prepared claims and
rewrites the whole file on every Prepare/Unprepare
(``cmd/nvidia-dra-plugin/checkpoint.go``, ``device_state.go:120-123,154,185``)
— a serialization bottleneck SURVEY.md §5.4 calls out against the
pods-scheduled/sec metric. Here each claim is its own checksummed file:

    <dir>/claims/<uid>.json   {"checksum": <crc32>, "v1": {...}}

so concurrent claim preparations never contend on one file, a claim write is
O(1) in total prepared claims, and recovery is a directory scan. Checksums
are CRC32 over the canonical JSON with the checksum field zeroed (same
construction as the reference's checkpoint.go:28-53).
"""

from __future__ import annotations

import json
import os
import zlib
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional

from ..utils.atomicfile import atomic_write_json, read_json


class CheckpointCorrupt(RuntimeError):
    pass


@dataclass
class PreparedDevice:
    """One prepared device of a claim, as returned to kubelet."""

    request_names: List[str]
    pool_name: str
    device_name: str
    cdi_device_ids: List[str]
    #: bookkeeping for unprepare
    parent_gpu_index: int = -1
    kind: str = "gpu"  # gpu | partition
    device_uuid: str = ""
    #: adminAccess results don't hold the GPU (no drain/exclusivity)
    admin: bool = False


@dataclass
class PreparedClaim:
    claim_uid: str
    namespace: str = ""
    name: str = ""
    devices: List[PreparedDevice] = field(default_factory=list)
    #: sharing bookkeeping for unprepare
    sharing_strategy: str = ""
    timeslice_gpus: List[int] = field(default_factory=list)
    shared_session_id: str = ""
    #: partition bookkeeping: gpu_index -> (compute, memory) applied
    repartitioned: Dict[str, List[str]] = field(default_factory=dict)
    #: claim-level CDI edits, persisted so the claim spec file can be
    #: regenerated if a crash lands between the two hot-path writes
    claim_env: List[str] = field(default_factory=list)
    claim_mounts: List[dict] = field(default_factory=list)

    def to_v1(self) -> dict:
        return asdict(self)

    @classmethod
    def from_v1(cls, d: dict) -> "PreparedClaim":
        devs = [PreparedDevice(**pd) for pd in d.pop("devices", [])]
        return cls(devices=devs, **{k: v for k, v in d.items()})


def _checksum(payload: dict) -> int:
    data = json.dumps(payload, sort_keys=True, separators=(",", ":")).encode()
    return zlib.crc32(data) & 0xFFFFFFFF


class CheckpointStore:
    """Per-claim checkpoint files under ``<root>/claims``."""

    def __init__(self, root: str):
        self.root = root
        self.claims_dir = os.path.join(root, "claims")
        os.makedirs(self.claims_dir, exist_ok=True)

    def _path(self, claim_uid: str) -> str:
        safe = claim_uid.replace("/", "_")
        return os.path.join(self.claims_dir, f"{safe}.json")

    def write(self, claim: PreparedClaim) -> None:
        v1 = claim.to_v1()
        atomic_write_json(
            self._path(claim.claim_uid), {"checksum": _checksum(v1), "v1": v1}
        )

    def read(self, claim_uid: str) -> Optional[PreparedClaim]:
        """None if absent; CheckpointCorrupt on checksum mismatch."""
        path = self._path(claim_uid)
        try:
            obj = read_json(path)
        except FileNotFoundError:
            return None
        except (json.JSONDecodeError, OSError) as e:
            raise CheckpointCorrupt(f"{path}: unreadable: {e}") from e
        v1 = obj.get("v1")
        if v1 is None or obj.get("checksum") != _checksum(v1):
            raise CheckpointCorrupt(f"{path}: checksum mismatch")
        return PreparedClaim.from_v1(v1)

    def delete(self, claim_uid: str) -> None:
        try:
            os.unlink(self._path(claim_uid))
        except FileNotFoundError:
            pass

    def list_all(self) -> Dict[str, PreparedClaim]:
        """Recovery scan; corrupt entries are surfaced, not skipped."""
        out: Dict[str, PreparedClaim] = {}
        for name in sorted(os.listdir(self.claims_dir)):
            if not name.endswith(".json") or name.startswith(".tmp-"):
                continue
            uid = name[: -len(".json")]
            claim = self.read(uid)
            if claim is not None:
                out[claim.claim_uid] = claim
        return out
