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
import threading
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
        import dataclasses

        dev_fields = {f.name for f in dataclasses.fields(PreparedDevice)}
        devs = [
            PreparedDevice(**{k: v for k, v in pd.items() if k in dev_fields})
            for pd in d.pop("devices", [])
        ]
        known = {f.name for f in dataclasses.fields(cls)}
        # unknown fields (written by a newer same-schema plugin) are
        # dropped rather than fatal — forward-compatible reads
        return cls(devices=devs, **{k: v for k, v in d.items() if k in known})


def _checksum(payload: dict) -> int:
    data = json.dumps(payload, sort_keys=True, separators=(",", ":")).encode()
    return zlib.crc32(data) & 0xFFFFFFFF


class CheckpointStore:
    """Per-claim checkpoint files under ``<root>/claims``.

    A write-through in-memory cache backs ``read``: the disk file is the
    durability record (fsynced), the cache is the hot-path record — the
    prepare idempotency lookup and unprepare both read the claim right
    back, and a disk read per operation was ~3% of lifecycle CPU on the
    MI355X pool (profiles/round2_hardware_notes.md). Cache misses fall
    through to disk (fresh store after restart), and disk reads still
    checksum-validate."""

    def __init__(self, root: str):
        self.root = root
        self.claims_dir = os.path.join(root, "claims")
        os.makedirs(self.claims_dir, exist_ok=True)
        self._lock = threading.Lock()
        self._cache: Dict[str, PreparedClaim] = {}
        #: uids known to have no checkpoint (negative cache for the
        #: idempotency miss on every fresh prepare)
        self._absent: set = set()

    def _path(self, claim_uid: str) -> str:
        safe = claim_uid.replace("/", "_")
        return os.path.join(self.claims_dir, f"{safe}.json")

    #: on-disk schema version; bump only with a migration in _read_disk
    SCHEMA_VERSION = 1

    def write(self, claim: PreparedClaim) -> None:
        v1 = claim.to_v1()
        atomic_write_json(
            self._path(claim.claim_uid),
            {
                "version": self.SCHEMA_VERSION,
                "checksum": _checksum(v1),
                "v1": v1,
            },
        )
        with self._lock:
            self._cache[claim.claim_uid] = claim
            self._absent.discard(claim.claim_uid)

    def read(self, claim_uid: str) -> Optional[PreparedClaim]:
        """None if absent; CheckpointCorrupt on checksum mismatch."""
        with self._lock:
            cached = self._cache.get(claim_uid)
            if cached is not None:
                return cached
            if claim_uid in self._absent:
                return None
        claim = self._read_disk(claim_uid)
        with self._lock:
            if claim is None:
                self._absent.add(claim_uid)
                if len(self._absent) > 100_000:  # bounded negative cache
                    self._absent.clear()
            else:
                self._cache[claim_uid] = claim
        return claim

    def _read_disk(self, claim_uid: str) -> Optional[PreparedClaim]:
        path = self._path(claim_uid)
        try:
            obj = read_json(path)
        except FileNotFoundError:
            return None
        except (json.JSONDecodeError, OSError) as e:
            raise CheckpointCorrupt(f"{path}: unreadable: {e}") from e
        version = obj.get("version", 1)  # round-1 files carry no field
        if version > self.SCHEMA_VERSION:
            raise CheckpointCorrupt(
                f"{path}: checkpoint schema v{version} is newer than this "
                f"plugin understands (v{self.SCHEMA_VERSION}) — refusing to "
                "guess (downgrade rollback safety)"
            )
        v1 = obj.get("v1")
        if v1 is None or obj.get("checksum") != _checksum(v1):
            raise CheckpointCorrupt(f"{path}: checksum mismatch")
        return PreparedClaim.from_v1(v1)

    def delete(self, claim_uid: str) -> None:
        with self._lock:
            self._cache.pop(claim_uid, None)
            self._absent.add(claim_uid)
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

    def invalidate_cache(self) -> None:
        """Drop the in-memory view (tests / external file manipulation)."""
        with self._lock:
            self._cache.clear()
            self._absent.clear()
