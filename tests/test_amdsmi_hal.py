"""AmdSmiDeviceLib logic tests with a mocked _amdhal extension:
processor->GPU grouping, partition sub-devices, arch decoding, xGMI link
attachment. These paths only run on multi-GPU / partitioned hardware, so
they are pinned here against realistic amdsmi output shapes (field names
match cpp/amdhal.cpp::describe_processor)."""

from typing import List


from k8s_dra_driver_amd.hal.amdsmi import AmdSmiDeviceLib


class FakeExt:
    """Mimics the _amdhal pybind surface."""

    def __init__(self, procs: List[dict]):
        self.procs = procs
        self.calls = []

    def init(self):
        self.calls.append("init")

    def shutdown(self):
        self.calls.append("shutdown")

    def reinit(self):
        self.calls.append("reinit")

    def enumerate(self):
        return [dict(p) for p in self.procs]

    def set_compute_partition(self, index, mode):
        self.calls.append(("set_compute", index, mode))

    def set_memory_partition(self, index, mode):
        self.calls.append(("set_memory", index, mode))


def proc(
    index,
    oam,
    *,
    uuid=None,
    bdf=None,
    partition_id=0,
    compute="SPX",
    memory="NPS1",
    kfd_node=None,
    links=None,
):
    return {
        "index": index,
        "uuid": uuid or f"uuid-gpu-{oam}",
        "market_name": "AMD Instinct MI355X",
        "product_name": "AMD Instinct MI355 OAM",
        "device_id": 0x75A0,
        "oam_id": oam,
        "num_compute_units": 256 if compute == "SPX" else 32,
        "target_graphics_version": 90500,
        "kfd_node_id": kfd_node if kfd_node is not None else oam + 1,
        "current_partition_id": partition_id,
        "vram_size_mb": 294896 if compute == "SPX" else 294896 // 8,
        "vram_type": 5,  # HBM3E
        "driver_version": "6.14.5",
        "bdf": bdf or f"0000:{0x10 + oam:02x}:00.0",
        "compute_partition": compute,
        "memory_partition": memory,
        "nps_caps": ["NPS1", "NPS2"],
        "xgmi_hive_id": 0x5A1AD,
        "xgmi_node_id": oam,
        "links": links or [],
    }


def make_lib(procs):
    lib = AmdSmiDeviceLib(sysfs_root="/nonexistent", ext=FakeExt(procs))
    lib.open()
    return lib


def test_eight_spx_gpus_grouped():
    procs = [proc(i, i) for i in range(8)]
    lib = make_lib(procs)
    gpus = lib.enumerate()
    assert len(gpus) == 8
    g0 = gpus[0]
    assert g0.architecture == "gfx950"
    assert g0.vram_total_mib == 294896
    assert g0.cu_count == 256
    assert g0.product_name == "AMD Instinct MI355 OAM"
    assert g0.vram_type == "HBM3E"
    assert g0.nps_caps == ["NPS1", "NPS2"]
    assert [g.oam_id for g in gpus] == list(range(8))


def test_cpx_processors_grouped_into_one_gpu():
    # one GPU in CPX: 8 amdsmi processors sharing oam 0
    procs = [
        proc(i, 0, partition_id=i, compute="CPX", kfd_node=i + 1)
        for i in range(8)
    ] + [proc(8, 1)]
    lib = make_lib(procs)
    gpus = lib.enumerate()
    assert len(gpus) == 2
    carved = next(g for g in gpus if g.oam_id == 0)
    whole = next(g for g in gpus if g.oam_id == 1)
    assert carved.compute_partition == "CPX"
    assert len(carved.partitions) == 8
    assert carved.cu_count == 8 * 32  # summed over partition processors
    assert carved.vram_total_mib == 294896 // 8 * 8
    assert [p.partition_id for p in carved.partitions] == list(range(8))
    assert carved.partitions[3].uuid == "uuid-gpu-0-cpx-3"
    assert whole.partitions == []


def test_xgmi_links_attached_by_bdf():
    l01 = {"bdf": "0000:11:00.0", "link_type": "XGMI", "max_bandwidth_gbs": 153, "bit_rate_gbs": 32}
    l10 = {"bdf": "0000:10:00.0", "link_type": "XGMI", "max_bandwidth_gbs": 153, "bit_rate_gbs": 32}
    pcie = {"bdf": "0000:00:01.0", "link_type": "PCIE", "max_bandwidth_gbs": 63, "bit_rate_gbs": 32}
    procs = [
        proc(0, 0, links=[l01, pcie]),
        proc(1, 1, links=[l10]),
    ]
    lib = make_lib(procs)
    gpus = lib.enumerate()
    assert gpus[0].xgmi_peer_oam_ids() == [1]
    assert gpus[1].xgmi_peer_oam_ids() == [0]
    assert gpus[0].links[0].max_bandwidth_gbps == 153


def test_set_partition_reinits_processor_list():
    procs = [proc(i, i) for i in range(2)]
    lib = make_lib(procs)
    lib.set_compute_partition(1, "CPX")
    ext = lib._ext
    assert ("set_compute", 1, "CPX") in ext.calls
    assert "reinit" in ext.calls


def test_head_proc_index_targets_partition_zero():
    # CPX GPU's processors listed out of order: head must be partition 0
    procs = [
        proc(0, 0, partition_id=3, compute="CPX"),
        proc(1, 0, partition_id=0, compute="CPX"),
        proc(2, 0, partition_id=1, compute="CPX"),
    ]
    lib = make_lib(procs)
    assert lib._head_proc_index(0) == 1
