"""gfx950 (MI355X) compute/memory partition catalog.

This is the MIG-profile analog. Where the reference enumerates every MIG
GPU-instance x compute-instance profile and its placements per GPU
(reference ``cmd/nvidia-dra-plugin/nvlib.go:244-295``), the MI355X has a much
simpler, whole-GPU partition model driven through amd-smi:

- **Compute partition** (``amdsmi_set_gpu_compute_partition``): SPX / DPX /
  QPX / CPX carve the 8 XCDs of one MI355X into 1 / 2 / 4 / 8 logical KFD
  devices. Each logical device appears as its own DRM render node.
- **Memory partition (NPS)** (``amdsmi_set_gpu_memory_partition``): NPS1 /
  NPS4 interleave the 8 HBM3E stacks as 1 or 4 NUMA domains.

Unlike MIG (per-instance profiles with free placement), a partition mode
applies to the *whole GPU*: the "profile" of a partitioned device is fully
determined by (compute mode, memory mode) and the chip constants. The
placements of a profile are the partition slots 0..N-1.

The reference's dynamic MIG create/delete was shipped commented-out
(``nvlib.go:560-669``); on MI355X dynamic repartition is real and implemented
in :mod:`k8s_dra_driver_amd.partition.manager`.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Sequence, Tuple

# Chip constants for AMD Instinct MI355X (gfx950).
GFX950_XCD_COUNT = 8
GFX950_CU_COUNT = 256
GFX950_CU_PER_XCD = GFX950_CU_COUNT // GFX950_XCD_COUNT  # 32
GFX950_VRAM_MIB = 288 * 1024  # 288 GiB HBM3E
GFX950_HBM_STACKS = 8
GFX950_XGMI_LINKS_PER_GPU = 7  # point-to-point, ~153 GB/s each
GFX950_ARCH = "gfx950"

#: compute partition mode -> number of logical devices
COMPUTE_MODES: Dict[str, int] = {
    "SPX": 1,  # Single Partition X-celerator: all 8 XCDs, one device
    "DPX": 2,  # Dual: 2 devices x 4 XCDs
    "QPX": 4,  # Quad: 4 devices x 2 XCDs
    "CPX": 8,  # Core: 8 devices x 1 XCD
}

#: memory partition (NPS) mode -> number of NUMA/memory domains
MEMORY_MODES: Dict[str, int] = {
    "NPS1": 1,
    "NPS2": 2,
    "NPS4": 4,
}

#: Default validity matrix for gfx950: compute mode -> allowed NPS modes.
#: A compute partition may never span multiple memory domains, i.e.
#: num_domains must divide num_partitions (or equal 1). Live hardware caps
#: (amdsmi_get_gpu_memory_partition_config nps_cap_mask) override this.
#: Measured round 1: the MI355X pool SKU reports compute caps
#: SPX/DPX/QPX/CPX but memory caps NPS1,NPS2 only (amd-smi partition) —
#: the MI300X-style NPS4 rows below are architectural upper bounds that
#: the live-caps override prunes on such firmware.
DEFAULT_VALID_NPS: Dict[str, Tuple[str, ...]] = {
    "SPX": ("NPS1",),
    "DPX": ("NPS1", "NPS2"),
    "QPX": ("NPS1", "NPS4"),
    "CPX": ("NPS1", "NPS4"),
}


@dataclass(frozen=True)
class PartitionProfile:
    """One (compute mode, memory mode) profile of a gfx950 GPU.

    The MIG-profile analog (reference ``nvlib.go:244-295`` /
    ``deviceinfo.go:144-206``): carries everything needed to publish the
    resulting logical devices as ResourceSlice entries, including the
    memory-slice capacity model the reference used for MIG placements
    (``deviceinfo.go:199-204``).
    """

    compute_mode: str  # SPX | DPX | QPX | CPX
    memory_mode: str  # NPS1 | NPS2 | NPS4
    num_partitions: int
    xcds_per_partition: int
    cus_per_partition: int
    memory_mib_per_partition: int
    num_memory_domains: int

    @property
    def name(self) -> str:
        return f"{self.compute_mode.lower()}-{self.memory_mode.lower()}"

    def memory_domain_of(self, partition_id: int) -> int:
        """NUMA/memory domain a compute partition is bound to.

        With P partitions over D domains (D divides P), partitions are
        packed contiguously: partition p -> domain p * D // P.
        """
        if not 0 <= partition_id < self.num_partitions:
            raise ValueError(
                f"partition_id {partition_id} out of range for {self.name}"
            )
        return partition_id * self.num_memory_domains // self.num_partitions

    def memory_slices_of(self, partition_id: int) -> List[int]:
        """HBM-stack slices claimed by a partition (memorySliceN capacities).

        The slice space has GFX950_HBM_STACKS entries regardless of NPS mode;
        a partition owns the contiguous run of stacks behind its share. This
        mirrors the reference's MIG memory-slice capacity model
        (``deviceinfo.go:199-204``) so CEL/constraint logic carries over.
        """
        per = GFX950_HBM_STACKS // self.num_partitions
        if per == 0:
            per = 1
        start = partition_id * GFX950_HBM_STACKS // self.num_partitions
        return list(range(start, start + per))


def validate_mode_combo(
    compute_mode: str,
    memory_mode: str,
    nps_caps: Sequence[str] | None = None,
) -> None:
    """Raise ValueError unless (compute, memory) is a valid gfx950 combo.

    ``nps_caps``: live capability list from hardware; defaults to the static
    gfx950 matrix.
    """
    if compute_mode not in COMPUTE_MODES:
        raise ValueError(
            f"unknown compute partition mode {compute_mode!r}; "
            f"expected one of {sorted(COMPUTE_MODES)}"
        )
    if memory_mode not in MEMORY_MODES:
        raise ValueError(
            f"unknown memory partition (NPS) mode {memory_mode!r}; "
            f"expected one of {sorted(MEMORY_MODES)}"
        )
    allowed = tuple(nps_caps) if nps_caps is not None else DEFAULT_VALID_NPS[compute_mode]
    if memory_mode not in allowed:
        raise ValueError(
            f"memory mode {memory_mode} not valid with compute mode "
            f"{compute_mode} (allowed: {allowed})"
        )
    # Structural invariant: a compute partition may not span memory domains.
    p = COMPUTE_MODES[compute_mode]
    d = MEMORY_MODES[memory_mode]
    if d > p or p % d != 0:
        raise ValueError(
            f"{compute_mode}x{memory_mode}: {d} memory domains do not pack "
            f"into {p} compute partitions"
        )


def make_profile(
    compute_mode: str,
    memory_mode: str,
    *,
    vram_total_mib: int = GFX950_VRAM_MIB,
    cu_count: int = GFX950_CU_COUNT,
    xcd_count: int = GFX950_XCD_COUNT,
    nps_caps: Sequence[str] | None = None,
) -> PartitionProfile:
    validate_mode_combo(compute_mode, memory_mode, nps_caps)
    p = COMPUTE_MODES[compute_mode]
    d = MEMORY_MODES[memory_mode]
    return PartitionProfile(
        compute_mode=compute_mode,
        memory_mode=memory_mode,
        num_partitions=p,
        xcds_per_partition=xcd_count // p,
        cus_per_partition=cu_count // p,
        # Guaranteed share: the partition's fraction of total VRAM. Under
        # e.g. CPX+NPS4 two partitions share each 72 GiB domain; capacity is
        # the per-partition guarantee (36 GiB), the domain id is published as
        # an attribute for co-scheduling.
        memory_mib_per_partition=vram_total_mib // p,
        num_memory_domains=d,
    )


def gfx950_catalog(
    *,
    vram_total_mib: int = GFX950_VRAM_MIB,
    cu_count: int = GFX950_CU_COUNT,
    xcd_count: int = GFX950_XCD_COUNT,
    nps_caps_by_mode: Dict[str, Sequence[str]] | None = None,
) -> List[PartitionProfile]:
    """Every valid (compute, NPS) profile for one gfx950 GPU."""
    out: List[PartitionProfile] = []
    for cm in COMPUTE_MODES:
        allowed = (
            nps_caps_by_mode.get(cm, DEFAULT_VALID_NPS[cm])
            if nps_caps_by_mode
            else DEFAULT_VALID_NPS[cm]
        )
        for mm in allowed:
            out.append(
                make_profile(
                    cm,
                    mm,
                    vram_total_mib=vram_total_mib,
                    cu_count=cu_count,
                    xcd_count=xcd_count,
                    nps_caps=allowed,
                )
            )
    return out


def profiles_for(compute_mode: str, **kw) -> List[PartitionProfile]:
    """All profiles with the given compute mode."""
    return [p for p in gfx950_catalog(**kw) if p.compute_mode == compute_mode]


def preferred_memory_mode(compute_mode: str, nps_caps=None) -> str:
    """Default NPS mode for a carve when the claim expresses no memory
    intent (scheduler-driven auto-carve): the highest NPS valid for the
    mode and reported by live caps — maximum locality per partition."""
    valid = DEFAULT_VALID_NPS.get(compute_mode, ("NPS1",))
    allowed = [m for m in valid if nps_caps is None or m in nps_caps]
    return allowed[-1] if allowed else "NPS1"
