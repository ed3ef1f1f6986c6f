"""Config API tests: strict decode, normalize/validate, precedence merge,
shared-compute memory-limit normalization (reference sharing_test.go parity).
"""

import pytest

from k8s_dra_driver_amd.api.types import (
    API_GROUP_VERSION,
    GpuConfig,
    GpuSharing,
    OpaqueConfig,
    PartitionConfig,
    SharedComputeSettings,
    StrictDecodeError,
    UnknownKindError,
    ValidationError,
    decode_config,
    parse_quantity_bytes,
    select_config_for_request,
)


def test_decode_gpu_config_minimal():
    cfg = decode_config({"apiVersion": API_GROUP_VERSION, "kind": "GpuConfig"})
    assert isinstance(cfg, GpuConfig)
    cfg.normalize()
    assert cfg.sharing.strategy == "TimeSlicing"
    assert cfg.sharing.time_slicing.interval == "Default"
    cfg.validate()


def test_decode_time_slicing_interval():
    cfg = decode_config(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": "TimeSlicing",
                "timeSlicingConfig": {"interval": "Long"},
            },
        }
    ).normalize()
    cfg.validate()
    assert cfg.sharing.time_slicing.quantum_us == 10000


def test_decode_shared_compute():
    cfg = decode_config(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {
                "strategy": "SharedCompute",
                "sharedComputeConfig": {
                    "defaultMemoryLimit": "8Gi",
                    "memoryLimits": {"0": "4Gi"},
                    "defaultCuSharePercent": 25,
                },
            },
        }
    ).normalize()
    cfg.validate()
    assert cfg.sharing.strategy == "SharedCompute"


def test_strict_decode_rejects_unknown_fields():
    with pytest.raises(StrictDecodeError):
        decode_config(
            {"apiVersion": API_GROUP_VERSION, "kind": "GpuConfig", "bogus": 1}
        )
    with pytest.raises(StrictDecodeError):
        decode_config(
            {
                "apiVersion": API_GROUP_VERSION,
                "kind": "GpuConfig",
                "sharing": {"strategy": "TimeSlicing", "mps": {}},
            }
        )


def test_unknown_kind_and_group():
    with pytest.raises(UnknownKindError):
        decode_config({"apiVersion": API_GROUP_VERSION, "kind": "Nope"})
    with pytest.raises(UnknownKindError):
        decode_config({"apiVersion": "gpu.nvidia.com/v1alpha1", "kind": "GpuConfig"})


def test_validation_errors():
    cfg = GpuConfig(sharing=GpuSharing(strategy="Bogus"))
    with pytest.raises(ValidationError):
        cfg.validate()
    bad_interval = decode_config(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "GpuConfig",
            "sharing": {"timeSlicingConfig": {"interval": "Tiny"}},
        }
    )
    with pytest.raises(ValidationError):
        bad_interval.validate()


def test_partition_config_decode_and_validate():
    cfg = decode_config(
        {
            "apiVersion": API_GROUP_VERSION,
            "kind": "PartitionConfig",
            "computePartition": "cpx",
            "memoryPartition": "nps4",
            "allowDynamicRepartition": True,
        }
    ).normalize()
    cfg.validate()
    assert cfg.compute_partition == "CPX"
    bad = PartitionConfig(compute_partition="SPX", memory_partition="NPS4")
    with pytest.raises(ValidationError):
        bad.validate()


def test_quantity_parsing():
    assert parse_quantity_bytes("4Gi") == 4 * 2**30
    assert parse_quantity_bytes("512M") == 512 * 10**6
    assert parse_quantity_bytes("1024") == 1024
    with pytest.raises(ValidationError):
        parse_quantity_bytes("four gigs")


class TestSharedComputeMemoryLimitNormalize:
    """Parity with reference sharing_test.go:28-160 (14-case table)."""

    uuids = {0: "uuid-a", 1: "uuid-b"}

    def test_default_applies_to_all(self):
        s = SharedComputeSettings(default_memory_limit="1Gi")
        out = s.normalized_memory_limits(self.uuids)
        assert out == {"uuid-a": 2**30, "uuid-b": 2**30}

    def test_index_resolution_and_override(self):
        s = SharedComputeSettings(
            default_memory_limit="1Gi", memory_limits={"1": "2Gi"}
        )
        out = s.normalized_memory_limits(self.uuids)
        assert out["uuid-a"] == 2**30
        assert out["uuid-b"] == 2 * 2**30

    def test_uuid_key_accepted(self):
        s = SharedComputeSettings(memory_limits={"uuid-a": "512Mi"})
        assert s.normalized_memory_limits(self.uuids) == {"uuid-a": 512 * 2**20}

    def test_unknown_index_rejected(self):
        s = SharedComputeSettings(memory_limits={"7": "1Gi"})
        with pytest.raises(ValidationError, match="index 7"):
            s.normalized_memory_limits(self.uuids)

    def test_unknown_uuid_rejected(self):
        s = SharedComputeSettings(memory_limits={"uuid-zz": "1Gi"})
        with pytest.raises(ValidationError, match="unknown device"):
            s.normalized_memory_limits(self.uuids)


def test_precedence_merge():
    """Reference device_state.go:225-259: claim > class > default; later
    entries win within a source; empty requests = all requests."""
    configs = [
        OpaqueConfig("default", [], GpuConfig().normalize()),
        OpaqueConfig(
            "class", [], GpuConfig(sharing=GpuSharing(strategy="SharedCompute"))
        ),
        OpaqueConfig(
            "claim",
            ["req-a"],
            GpuConfig(
                sharing=GpuSharing(
                    strategy="TimeSlicing",
                )
            ),
        ),
    ]
    got_a = select_config_for_request("req-a", configs, GpuConfig)
    assert got_a.sharing.strategy == "TimeSlicing"  # claim wins for req-a
    got_b = select_config_for_request("req-b", configs, GpuConfig)
    assert got_b.sharing.strategy == "SharedCompute"  # class wins for req-b
    # a request with only the default
    got_c = select_config_for_request("req-b", configs[:1], GpuConfig)
    assert got_c.sharing.strategy == "TimeSlicing"
    # kind filtering
    assert select_config_for_request("req-a", configs, PartitionConfig) is None
