"""Checkpoint tests: per-claim files, checksums, recovery scan."""

import json
import os

import pytest

from k8s_dra_driver_amd.state.checkpoint import (
    CheckpointCorrupt,
    CheckpointStore,
    PreparedClaim,
    PreparedDevice,
)


def _claim(uid="uid-1"):
    return PreparedClaim(
        claim_uid=uid,
        namespace="default",
        name="claim-a",
        devices=[
            PreparedDevice(
                request_names=["gpu"],
                pool_name="node-1",
                device_name="gpu-0",
                cdi_device_ids=["k8s.gpu.amd.com/device=gpu-0"],
                parent_gpu_index=0,
                kind="gpu",
                device_uuid="amd-mi355x-00",
            )
        ],
        sharing_strategy="TimeSlicing",
        timeslice_gpus=[0],
    )


def test_roundtrip(tmp_path):
    store = CheckpointStore(str(tmp_path))
    store.write(_claim())
    got = store.read("uid-1")
    assert got.claim_uid == "uid-1"
    assert got.devices[0].device_name == "gpu-0"
    assert got.timeslice_gpus == [0]


def test_absent_returns_none(tmp_path):
    assert CheckpointStore(str(tmp_path)).read("nope") is None


def test_checksum_detects_tamper(tmp_path):
    store = CheckpointStore(str(tmp_path))
    store.write(_claim())
    path = os.path.join(str(tmp_path), "claims", "uid-1.json")
    obj = json.load(open(path))
    obj["v1"]["devices"][0]["device_name"] = "gpu-7"
    json.dump(obj, open(path, "w"))
    # a restarted store (fresh cache) must reject the tampered file
    with pytest.raises(CheckpointCorrupt, match="checksum"):
        CheckpointStore(str(tmp_path)).read("uid-1")
    # and an explicit cache invalidation hits the disk copy too
    store.invalidate_cache()
    with pytest.raises(CheckpointCorrupt, match="checksum"):
        store.read("uid-1")


def test_cache_survives_lifecycle_and_restart_reads_disk(tmp_path):
    store = CheckpointStore(str(tmp_path))
    store.write(_claim())
    # hot-path read is served (from cache) and equals the written claim
    pc = store.read("uid-1")
    assert pc.devices[0].device_name == "gpu-0"
    store.delete("uid-1")
    assert store.read("uid-1") is None  # negative cache
    # fresh store after restart agrees with disk
    assert CheckpointStore(str(tmp_path)).read("uid-1") is None


def test_per_claim_files_are_independent(tmp_path):
    store = CheckpointStore(str(tmp_path))
    store.write(_claim("a"))
    store.write(_claim("b"))
    files = os.listdir(tmp_path / "claims")
    assert sorted(files) == ["a.json", "b.json"]
    store.delete("a")
    assert store.read("a") is None
    assert store.read("b") is not None


def test_recovery_scan(tmp_path):
    store = CheckpointStore(str(tmp_path))
    for uid in ("a", "b", "c"):
        store.write(_claim(uid))
    fresh = CheckpointStore(str(tmp_path))
    recovered = fresh.list_all()
    assert sorted(recovered) == ["a", "b", "c"]
    assert recovered["b"].devices[0].pool_name == "node-1"


def test_schema_versioning(tmp_path):
    """Upgrade safety: round-1 files (no version field) read fine; files
    from a NEWER schema are refused loudly; unknown fields from a newer
    same-schema plugin are tolerated."""
    store = CheckpointStore(str(tmp_path))
    store.write(_claim())
    path = os.path.join(str(tmp_path), "claims", "uid-1.json")
    obj = json.load(open(path))
    assert obj["version"] == CheckpointStore.SCHEMA_VERSION

    # round-1 file: no version field -> treated as v1
    del obj["version"]
    json.dump(obj, open(path, "w"))
    assert CheckpointStore(str(tmp_path)).read("uid-1") is not None

    # future schema -> loud refusal
    obj["version"] = 99
    json.dump(obj, open(path, "w"))
    with pytest.raises(CheckpointCorrupt, match="newer than this plugin"):
        CheckpointStore(str(tmp_path)).read("uid-1")


def test_unknown_fields_tolerated(tmp_path):
    store = CheckpointStore(str(tmp_path))
    store.write(_claim())
    path = os.path.join(str(tmp_path), "claims", "uid-1.json")
    obj = json.load(open(path))
    # a newer plugin added fields within the same schema version
    obj["v1"]["future_field"] = {"x": 1}
    obj["v1"]["devices"][0]["future_dev_field"] = True
    obj["checksum"] = None  # recompute below
    import zlib as _zlib

    data = json.dumps(obj["v1"], sort_keys=True, separators=(",", ":")).encode()
    obj["checksum"] = _zlib.crc32(data) & 0xFFFFFFFF
    json.dump(obj, open(path, "w"))
    pc = CheckpointStore(str(tmp_path)).read("uid-1")
    assert pc is not None and pc.devices[0].device_name == "gpu-0"
