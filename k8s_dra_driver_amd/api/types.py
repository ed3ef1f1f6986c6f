"""Opaque per-claim configuration API (group ``resource.gpu.amd.com/v1alpha1``).

The analog of the reference's ``api/nvidia.com/resource/gpu/v1alpha1``
(GpuConfig / MigDeviceConfig / ImexChannelConfig): these are **not CRDs** —
they are opaque parameter payloads embedded in DeviceClass / ResourceClaim
``config.opaque.parameters`` and strict-decoded by the kubelet plugin
(reference ``api.go:43-71``, ``device_state.go:457-510``).

Kinds:

- ``GpuConfig`` — sharing strategy for whole-GPU / partition claims:
  ``TimeSlicing`` (amdgpu scheduler quantum; reference sharing.go:97-122) or
  ``SharedCompute`` (the MPS analog: a node-local supervisor scoping shm +
  CU shares — no control-daemon Deployment).
- ``PartitionConfig`` — the MigDeviceConfig analog, plus *dynamic*
  repartition intent (possible on MI355X, disabled in the reference,
  ``nvlib.go:560-669``): requests the parent GPU be switched to a
  compute/NPS mode as part of Prepare.

Decoding is strict: unknown group/kind/fields are errors (reference uses a
strict JSON serializer, ``api.go:63-70``).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..partition.catalog import validate_mode_combo

API_GROUP = "resource.gpu.amd.com"
API_VERSION = "v1alpha1"
API_GROUP_VERSION = f"{API_GROUP}/{API_VERSION}"

# Sharing strategies
TIME_SLICING = "TimeSlicing"
SHARED_COMPUTE = "SharedCompute"  # MPS analog
SHARING_STRATEGIES = (TIME_SLICING, SHARED_COMPUTE)

# Time-slice intervals (reference: Default/Short/Medium/Long,
# sharing.go:97-122) mapped to amdgpu scheduler quanta in microseconds.
TIMESLICE_INTERVALS: Dict[str, Optional[int]] = {
    "Default": None,  # leave the scheduler default
    "Short": 1000,
    "Medium": 3000,
    "Long": 10000,
}


class ConfigError(ValueError):
    """Sentinel-style base error for config decode/validation failures."""


class UnknownKindError(ConfigError):
    pass


class StrictDecodeError(ConfigError):
    pass


class ValidationError(ConfigError):
    pass


_QUANTITY_RE = re.compile(r"^([0-9]+(?:\.[0-9]+)?)\s*(Ki|Mi|Gi|Ti|K|M|G|T)?$")
_QUANTITY_MULT = {
    None: 1,
    "K": 10**3,
    "M": 10**6,
    "G": 10**9,
    "T": 10**12,
    "Ki": 2**10,
    "Mi": 2**20,
    "Gi": 2**30,
    "Ti": 2**40,
}


def parse_quantity_bytes(s: str) -> int:
    """Parse a k8s-style quantity into bytes (reference sharing.go:238-273
    does Mi->M conversion for MPS limits; we keep exact bytes)."""
    m = _QUANTITY_RE.match(str(s).strip())
    if not m:
        raise ValidationError(f"invalid quantity {s!r}")
    val, suffix = m.groups()
    return int(float(val) * _QUANTITY_MULT[suffix])


def _reject_unknown(obj: Dict[str, Any], allowed: set, where: str) -> None:
    unknown = set(obj) - allowed
    if unknown:
        raise StrictDecodeError(
            f"{where}: unknown field(s) {sorted(unknown)} (strict decoding)"
        )


# ---------------------------------------------------------------------------
# Sharing
# ---------------------------------------------------------------------------


@dataclass
class TimeSlicingSettings:
    interval: str = "Default"

    def validate(self) -> None:
        if self.interval not in TIMESLICE_INTERVALS:
            raise ValidationError(
                f"unknown time-slice interval {self.interval!r}; expected one "
                f"of {sorted(TIMESLICE_INTERVALS)}"
            )

    @property
    def quantum_us(self) -> Optional[int]:
        return TIMESLICE_INTERVALS[self.interval]


@dataclass
class SharedComputeSettings:
    """MPS-analog settings.

    ``default_memory_limit``: per-client VRAM cap (quantity string) applied
    to every device unless overridden; ``memory_limits`` keys may be device
    *indices* ("0") or UUIDs — normalized to UUIDs against the claim's
    devices, mirroring MpsPerDevicePinnedMemoryLimit.Normalize (reference
    api sharing.go:188-273).
    ``default_cu_share_percent``: active-CU share per client (the
    set_default_active_thread_percentage analog, applied via CU masking).
    """

    default_memory_limit: Optional[str] = None
    memory_limits: Dict[str, str] = field(default_factory=dict)
    default_cu_share_percent: Optional[int] = None

    def validate(self) -> None:
        if self.default_memory_limit is not None:
            parse_quantity_bytes(self.default_memory_limit)
        for k, v in self.memory_limits.items():
            parse_quantity_bytes(v)
        if self.default_cu_share_percent is not None and not (
            1 <= self.default_cu_share_percent <= 100
        ):
            raise ValidationError(
                f"cuSharePercent must be in [1,100], got {self.default_cu_share_percent}"
            )

    def normalized_memory_limits(self, device_uuids_by_index: Dict[int, str]) -> Dict[str, int]:
        """Resolve index keys to UUIDs and quantities to bytes.

        Rules (reference sharing_test.go:28-160 semantics):
        - an index key must resolve to a claim device, else error;
        - a UUID key must belong to the claim, else error;
        - an explicit per-device limit overrides the default;
        - devices with no explicit limit get the default (if any).
        """
        out: Dict[str, int] = {}
        if self.default_memory_limit is not None:
            for uuid in device_uuids_by_index.values():
                out[uuid] = parse_quantity_bytes(self.default_memory_limit)
        known = set(device_uuids_by_index.values())
        for key, limit in self.memory_limits.items():
            if re.fullmatch(r"[0-9]+", key):
                idx = int(key)
                if idx not in device_uuids_by_index:
                    raise ValidationError(
                        f"memory limit references device index {idx} not in claim"
                    )
                uuid = device_uuids_by_index[idx]
            else:
                if key not in known:
                    raise ValidationError(
                        f"memory limit references unknown device {key!r}"
                    )
                uuid = key
            out[uuid] = parse_quantity_bytes(limit)
        return out


@dataclass
class GpuSharing:
    strategy: str = TIME_SLICING
    time_slicing: TimeSlicingSettings = field(default_factory=TimeSlicingSettings)
    shared_compute: SharedComputeSettings = field(default_factory=SharedComputeSettings)

    def validate(self) -> None:
        if self.strategy not in SHARING_STRATEGIES:
            raise ValidationError(
                f"unknown sharing strategy {self.strategy!r}; expected one of "
                f"{SHARING_STRATEGIES}"
            )
        self.time_slicing.validate()
        self.shared_compute.validate()


# ---------------------------------------------------------------------------
# Config kinds
# ---------------------------------------------------------------------------


@dataclass
class GpuConfig:
    """Per-claim GPU configuration (reference gpuconfig.go:29-67)."""

    sharing: Optional[GpuSharing] = None

    KIND = "GpuConfig"

    def normalize(self) -> "GpuConfig":
        """Fill defaults (reference Normalize, gpuconfig.go:52-67): an empty
        config means time-slicing at the Default interval."""
        if self.sharing is None:
            self.sharing = GpuSharing()
        return self

    def validate(self) -> None:
        if self.sharing is not None:
            self.sharing.validate()


@dataclass
class PartitionConfig:
    """Partition intent for the parent GPU (MigDeviceConfig analog,
    reference migconfig.go:29-64 — plus dynamic repartition, which the
    reference could not ship)."""

    compute_partition: str = "SPX"
    memory_partition: str = "NPS1"
    #: allow Prepare to switch the parent GPU's mode if needed (requires the
    #: GPU to be otherwise unallocated; partition.manager enforces).
    allow_dynamic_repartition: bool = False

    KIND = "PartitionConfig"

    def normalize(self) -> "PartitionConfig":
        self.compute_partition = self.compute_partition.upper() or "SPX"
        self.memory_partition = self.memory_partition.upper() or "NPS1"
        return self

    def validate(self) -> None:
        try:
            validate_mode_combo(self.compute_partition, self.memory_partition)
        except ValueError as e:
            raise ValidationError(str(e)) from e


# ---------------------------------------------------------------------------
# Strict decoder
# ---------------------------------------------------------------------------


def decode_config(obj: Dict[str, Any]):
    """Strict-decode one opaque parameters object into a config instance.

    The reference registers its scheme and uses a strict serializer
    (``api.go:45-71``); unknown apiVersion/kind/fields are errors here too.
    """
    if not isinstance(obj, dict):
        raise StrictDecodeError(f"opaque parameters must be an object, got {type(obj)}")
    api_version = obj.get("apiVersion")
    kind = obj.get("kind")
    if api_version != API_GROUP_VERSION:
        raise UnknownKindError(
            f"unknown apiVersion {api_version!r} (want {API_GROUP_VERSION})"
        )
    if kind == GpuConfig.KIND:
        _reject_unknown(obj, {"apiVersion", "kind", "sharing"}, "GpuConfig")
        sharing = None
        if "sharing" in obj and obj["sharing"] is not None:
            s = obj["sharing"]
            _reject_unknown(
                s,
                {"strategy", "timeSlicingConfig", "sharedComputeConfig"},
                "GpuConfig.sharing",
            )
            ts = TimeSlicingSettings()
            if s.get("timeSlicingConfig"):
                t = s["timeSlicingConfig"]
                _reject_unknown(t, {"interval"}, "timeSlicingConfig")
                ts = TimeSlicingSettings(interval=t.get("interval", "Default"))
            sc = SharedComputeSettings()
            if s.get("sharedComputeConfig"):
                c = s["sharedComputeConfig"]
                _reject_unknown(
                    c,
                    {
                        "defaultMemoryLimit",
                        "memoryLimits",
                        "defaultCuSharePercent",
                    },
                    "sharedComputeConfig",
                )
                sc = SharedComputeSettings(
                    default_memory_limit=c.get("defaultMemoryLimit"),
                    memory_limits=dict(c.get("memoryLimits") or {}),
                    default_cu_share_percent=c.get("defaultCuSharePercent"),
                )
            sharing = GpuSharing(
                strategy=s.get("strategy", TIME_SLICING),
                time_slicing=ts,
                shared_compute=sc,
            )
        return GpuConfig(sharing=sharing)
    if kind == PartitionConfig.KIND:
        _reject_unknown(
            obj,
            {
                "apiVersion",
                "kind",
                "computePartition",
                "memoryPartition",
                "allowDynamicRepartition",
            },
            "PartitionConfig",
        )
        return PartitionConfig(
            compute_partition=str(obj.get("computePartition", "SPX")),
            memory_partition=str(obj.get("memoryPartition", "NPS1")),
            allow_dynamic_repartition=bool(obj.get("allowDynamicRepartition", False)),
        )
    raise UnknownKindError(f"unknown kind {kind!r} in group {API_GROUP}")


# ---------------------------------------------------------------------------
# Precedence merge over a claim's allocation results
# ---------------------------------------------------------------------------


@dataclass
class OpaqueConfig:
    """One decoded config + where it came from + which requests it covers.

    Mirrors the reference's GetOpaqueDeviceConfigs output
    (``device_state.go:457-510``): source is "class" or "claim"; requests
    empty = applies to all requests.
    """

    source: str  # "default" | "class" | "claim"
    requests: List[str]
    config: Any


PRECEDENCE = {"default": 0, "class": 1, "claim": 2}


def select_config_for_request(
    request_name: str, configs: List[OpaqueConfig], want_kind: type
):
    """Highest-precedence config of ``want_kind`` covering ``request_name``.

    Reference semantics (``device_state.go:225-259``): defaults are
    prepended (lowest precedence); among applicable configs the *last* one
    of the highest source precedence wins; a config with an empty requests
    list applies to every request.
    """
    best = None
    best_rank = -1
    for i, oc in enumerate(configs):
        if not isinstance(oc.config, want_kind):
            continue
        if oc.requests and request_name not in oc.requests:
            continue
        rank = PRECEDENCE[oc.source] * len(configs) + i
        if rank >= best_rank:
            best, best_rank = oc, rank
    return best.config if best else None
