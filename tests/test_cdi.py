"""CDI generation tests: spec shape, atomicity, claim lifecycle."""

import json
import os

import pytest

from k8s_dra_driver_amd.cdi.handler import CDIHandler, CLAIM_KIND, DEVICE_KIND
from k8s_dra_driver_amd.cdi.spec import (
    CDISpec,
    CDIDevice,
    ContainerEdits,
    DeviceNode,
    Mount,
    read_spec_file,
    write_spec_file,
)
from k8s_dra_driver_amd.hal.model import AllocatableDevice


@pytest.fixture
def handler(tmp_path):
    return CDIHandler(cdi_root=str(tmp_path / "cdi"))


def _allocatable(lib):
    devs = []
    for g in lib.enumerate():
        if g.partitions:
            for p in g.partitions:
                devs.append(AllocatableDevice.from_partition(g, p))
        else:
            devs.append(AllocatableDevice.from_gpu(g))
    return devs


def test_base_spec_injects_kfd_and_render_nodes(handler, fake_lib):
    path = handler.create_standard_spec(_allocatable(fake_lib))
    spec = read_spec_file(path)
    assert spec["kind"] == DEVICE_KIND
    common = spec["containerEdits"]
    assert {"path": "/dev/kfd", "type": "c", "permissions": "rw"} in common["deviceNodes"]
    assert "AMD_VISIBLE_DEVICES=void" in common["env"]
    assert len(spec["devices"]) == 8
    gpu0 = next(d for d in spec["devices"] if d["name"] == "gpu-0")
    paths = [n["path"] for n in gpu0["containerEdits"]["deviceNodes"]]
    assert any(p.startswith("/dev/dri/renderD") for p in paths)
    # card nodes are included only when they exist on the host (this test
    # container has no /dev/dri) — renderD + kfd are the required pair.


def test_base_spec_partitioned_gpu(handler, fake_lib):
    fake_lib.set_compute_partition(0, "CPX")
    devs = _allocatable(fake_lib)
    path = handler.create_standard_spec(devs)
    spec = read_spec_file(path)
    names = [d["name"] for d in spec["devices"]]
    assert "gpu-0-cpx-0" in names and "gpu-0-cpx-7" in names
    assert "gpu-1" in names
    assert len(names) == 8 + 7  # 8 partitions + 7 whole GPUs
    # each partition injects a distinct render node
    renders = set()
    for d in spec["devices"]:
        if d["name"].startswith("gpu-0-cpx-"):
            renders.update(
                n["path"]
                for n in d["containerEdits"]["deviceNodes"]
                if "renderD" in n["path"]
            )
    assert len(renders) == 8


def test_claim_spec_lifecycle(handler):
    uid = "3f2c1a9e-claim"
    edits = ContainerEdits(env=["AMD_DRA_CLAIM=1"])
    path = handler.create_claim_spec(uid, ["gpu-0", "gpu-1"], edits)
    spec = read_spec_file(path)
    assert spec["kind"] == CLAIM_KIND
    assert [d["name"] for d in spec["devices"]] == [
        f"{uid}-gpu-0",
        f"{uid}-gpu-1",
    ]
    assert handler.list_claim_spec_uids() == [uid]
    handler.delete_claim_spec(uid)
    assert handler.list_claim_spec_uids() == []
    handler.delete_claim_spec(uid)  # idempotent


def test_cdi_ids(handler):
    assert handler.device_id("gpu-3") == "k8s.gpu.amd.com/device=gpu-3"
    assert (
        handler.claim_device_id("uid-1", "gpu-3")
        == "k8s.gpu.amd.com/claim=uid-1-gpu-3"
    )


def test_minimum_version_stamping(tmp_path):
    plain = CDISpec(
        kind=DEVICE_KIND,
        devices=[
            CDIDevice("d0", ContainerEdits(device_nodes=[DeviceNode("/dev/kfd")]))
        ],
    )
    assert plain.minimum_version() == "0.5.0"
    with_mounts = CDISpec(
        kind=CLAIM_KIND,
        devices=[
            CDIDevice(
                "d0",
                ContainerEdits(mounts=[Mount("/host/shm", "/dev/shm")]),
            )
        ],
    )
    assert with_mounts.minimum_version() == "0.6.0"


def test_atomic_write_leaves_no_tmp(tmp_path):
    spec = CDISpec(kind=DEVICE_KIND)
    target = str(tmp_path / "cdi" / "out.json")
    write_spec_file(spec, target)
    write_spec_file(spec, target)  # overwrite path
    files = os.listdir(tmp_path / "cdi")
    assert files == ["out.json"]
    assert json.load(open(target))["kind"] == DEVICE_KIND


def test_dev_root_prefix(tmp_path, fake_lib):
    h = CDIHandler(cdi_root=str(tmp_path), dev_root="/driver-root")
    path = h.create_standard_spec(_allocatable(fake_lib))
    spec = read_spec_file(path)
    kfd = spec["containerEdits"]["deviceNodes"][0]
    assert kfd["hostPath"] == "/driver-root/dev/kfd"
    assert kfd["path"] == "/dev/kfd"


def test_rocm_mount_option(tmp_path, fake_lib):
    h = CDIHandler(cdi_root=str(tmp_path), rocm_mount="/opt/rocm-7.2.0")
    path = h.create_standard_spec(_allocatable(fake_lib))
    spec = read_spec_file(path)
    common = spec["containerEdits"]
    mounts = common["mounts"]
    assert mounts[0]["hostPath"] == "/opt/rocm-7.2.0"
    assert mounts[0]["containerPath"] == "/opt/rocm"
    assert "ROCM_PATH=/opt/rocm" in common["env"]
    assert spec["cdiVersion"] == "0.6.0"  # mounts bump the min version
