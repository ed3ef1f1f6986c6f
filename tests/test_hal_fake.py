"""Fake HAL tests: enumeration, partition state machine, faults.

The fake backend is itself a deliverable (the reference has no
hardware-free backend) so its semantics are pinned here.
"""

import pytest

from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.base import HalError
from k8s_dra_driver_amd.hal.fake import FaultInjector
from k8s_dra_driver_amd.hal.model import (
    AllocatableDevice,
    gpu_to_device,
    partition_to_device,
)


def test_enumerate_eight_mi355x(fake_lib):
    gpus = fake_lib.enumerate()
    assert len(gpus) == 8
    g0 = gpus[0]
    assert g0.canonical_name == "gpu-0"
    assert g0.architecture == "gfx950"
    assert g0.vram_total_mib == 288 * 1024
    assert g0.cu_count == 256
    assert g0.compute_partition == "SPX"
    # full xGMI mesh: 7 p2p links per GPU
    assert all(len(g.links) == 7 for g in gpus)
    assert len({g.uuid for g in gpus}) == 8


def test_enumeration_is_a_snapshot(fake_lib):
    a = fake_lib.enumerate()[0]
    a.compute_partition = "CPX"  # mutate the copy
    b = fake_lib.enumerate()[0]
    assert b.compute_partition == "SPX"


def test_cpx_switch_exposes_eight_partitions(fake_lib):
    fake_lib.set_memory_partition(0, "NPS1")
    fake_lib.set_compute_partition(0, "CPX")
    g0 = fake_lib.enumerate()[0]
    assert g0.compute_partition == "CPX"
    assert len(g0.partitions) == 8
    names = [p.canonical_name for p in g0.partitions]
    assert names[0] == "gpu-0-cpx-0" and names[7] == "gpu-0-cpx-7"
    # each partition has its own render node; none collide with gpu-1's
    minors = {p.render_minor for p in g0.partitions}
    assert len(minors) == 8
    g1 = fake_lib.enumerate()[1]
    assert g1.render_minor not in minors


def test_partition_identity_stable_across_cycles(fake_lib):
    fake_lib.set_compute_partition(0, "CPX")
    first = [p.uuid for p in fake_lib.enumerate()[0].partitions]
    fake_lib.set_compute_partition(0, "SPX")
    fake_lib.set_compute_partition(0, "CPX")
    second = [p.uuid for p in fake_lib.enumerate()[0].partitions]
    assert first == second  # names/uuids keyed on (gpu uuid, mode, slot)


def test_nps_ordering_enforced(fake_lib):
    # NPS4 while SPX is invalid; CPX then NPS4 is the legal order
    with pytest.raises(HalError):
        fake_lib.set_memory_partition(0, "NPS4")
    fake_lib.set_compute_partition(0, "CPX")
    fake_lib.set_memory_partition(0, "NPS4")
    g0 = fake_lib.enumerate()[0]
    assert g0.memory_partition == "NPS4"
    # switching back to SPX under NPS4 must be refused
    with pytest.raises(HalError):
        fake_lib.set_compute_partition(0, "SPX")
    fake_lib.set_memory_partition(0, "NPS1")
    fake_lib.set_compute_partition(0, "SPX")


def test_busy_gpu_refuses_repartition(fake_lib):
    fake_lib.mark_busy(3, pid=4242)
    with pytest.raises(HalError, match="busy|using the GPU"):
        fake_lib.set_compute_partition(3, "CPX")
    fake_lib.mark_idle(3)
    fake_lib.set_compute_partition(3, "CPX")


def test_observer_fires_on_partition_change(fake_lib):
    events = []
    fake_lib.subscribe(lambda: events.append(1))
    fake_lib.set_compute_partition(0, "CPX")
    fake_lib.set_compute_partition(0, "CPX")  # no-op: no event
    assert len(events) == 1


def test_fault_injection():
    faults = FaultInjector()
    lib = FakeDeviceLib(faults=faults)
    lib.open()
    faults.fail_next("enumerate", HalError("injected"))
    with pytest.raises(HalError, match="injected"):
        lib.enumerate()
    assert len(lib.enumerate()) == 8  # next call succeeds
    assert faults.call_counts["enumerate"] == 2


def test_device_node_paths(fake_lib):
    paths = fake_lib.device_node_paths(0)
    assert paths["kfd"] == "/dev/kfd"
    assert paths["renderD"].startswith("/dev/dri/renderD")
    fake_lib.set_compute_partition(0, "CPX")
    p3 = fake_lib.device_node_paths(0, partition_id=3)
    assert p3["renderD"] != paths["renderD"]


def test_resourceslice_projection_gpu(fake_lib):
    g0 = fake_lib.enumerate()[0]
    dev = gpu_to_device(g0)
    assert dev["name"] == "gpu-0"
    attrs = dev["basic"]["attributes"]
    assert attrs["gpu.amd.com/type"] == {"string": "gpu"}
    assert attrs["gpu.amd.com/architecture"] == {"string": "gfx950"}
    assert attrs["gpu.amd.com/xgmiLinkCount"] == {"int": 7}
    caps = dev["basic"]["capacity"]
    assert caps["gpu.amd.com/memory"] == {"value": "288Gi"}


def test_resourceslice_projection_partition(fake_lib):
    fake_lib.set_compute_partition(0, "CPX")
    fake_lib.set_memory_partition(0, "NPS4")
    g0 = fake_lib.enumerate()[0]
    dev = partition_to_device(g0, g0.partitions[3])
    assert dev["name"] == "gpu-0-cpx-3"
    attrs = dev["basic"]["attributes"]
    assert attrs["gpu.amd.com/parentUUID"] == {"string": g0.uuid}
    assert attrs["gpu.amd.com/memoryDomain"] == {"int": 1}
    caps = dev["basic"]["capacity"]
    assert caps["gpu.amd.com/memory"] == {"value": "36Gi"}
    assert caps["gpu.amd.com/memorySlice3"] == {"value": "1"}


def test_allocatable_device_union(fake_lib):
    fake_lib.set_compute_partition(1, "DPX")
    gpus = fake_lib.enumerate()
    whole = AllocatableDevice.from_gpu(gpus[0])
    part = AllocatableDevice.from_partition(gpus[1], gpus[1].partitions[1])
    assert whole.kind == "gpu" and part.kind == "partition"
    assert part.canonical_name == "gpu-1-dpx-1"
    assert part.parent_gpu.uuid == gpus[1].uuid
    assert whole.to_device()["name"] == "gpu-0"


def test_dpx_nps2_memory_domains(fake_lib):
    """DPX+NPS2: 2 partitions over 2 domains, 1:1 binding."""
    fake_lib.set_compute_partition(2, "DPX")
    fake_lib.set_memory_partition(2, "NPS2")
    g = fake_lib.enumerate()[2]
    assert (g.compute_partition, g.memory_partition) == ("DPX", "NPS2")
    assert len(g.partitions) == 2
    prof = g.partitions[0].profile
    assert prof.num_memory_domains == 2
    assert [prof.memory_domain_of(i) for i in range(2)] == [0, 1]
    assert prof.memory_mib_per_partition == 288 * 1024 // 2
    from k8s_dra_driver_amd.hal.model import partition_to_device

    dev = partition_to_device(g, g.partitions[1])
    assert dev["basic"]["attributes"]["gpu.amd.com/memoryDomain"] == {"int": 1}
    assert dev["basic"]["capacity"]["gpu.amd.com/memory"] == {"value": "144Gi"}
