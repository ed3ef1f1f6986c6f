"""KFD sysfs topology parser tests against a fixture tree."""

import os

from k8s_dra_driver_amd.hal.sysfs import (
    KFD_IOLINK_TYPE_XGMI,
    KfdTopology,
)


def build_fixture(root, num_gpus=2):
    nodes = os.path.join(root, "class", "kfd", "kfd", "topology", "nodes")
    # node 0: CPU (no simds)
    os.makedirs(os.path.join(nodes, "0"))
    with open(os.path.join(nodes, "0", "properties"), "w") as f:
        f.write("cpu_cores_count 96\nsimd_count 0\n")
    for i in range(1, num_gpus + 1):
        d = os.path.join(nodes, str(i))
        os.makedirs(os.path.join(d, "io_links", "0"))
        with open(os.path.join(d, "properties"), "w") as f:
            f.write(
                f"simd_count 1024\nsimd_per_cu 4\n"
                f"gfx_target_version 90500\n"
                f"drm_render_minor {127 + i}\n"
                f"location_id {0x0300 + i * 0x100}\ndomain 0\n"
                f"unique_id {0xABC000 + i}\n"
            )
        # xGMI link to the other GPU node
        peer = 2 if i == 1 else 1
        with open(
            os.path.join(d, "io_links", "0", "properties"), "w"
        ) as f:
            f.write(
                f"type {KFD_IOLINK_TYPE_XGMI}\nnode_from {i}\nnode_to {peer}\n"
                f"weight 15\nmax_bandwidth 153000\n"
            )
    # drm card/render mapping
    drm = os.path.join(root, "class", "drm")
    for i in range(1, num_gpus + 1):
        dev = os.path.join(root, "devices", f"pci-gpu-{i}")
        os.makedirs(dev, exist_ok=True)
        for name in (f"card{i - 1}", f"renderD{127 + i}"):
            os.makedirs(os.path.join(drm, name), exist_ok=True)
            os.symlink(dev, os.path.join(drm, name, "device"))
    return root


def test_gpu_nodes_parsed(tmp_path):
    topo = KfdTopology(str(build_fixture(tmp_path)))
    assert topo.available()
    gpus = topo.gpu_nodes()
    assert len(gpus) == 2  # CPU node excluded
    g = gpus[0]
    assert g.gfx_arch == "gfx950"
    assert g.cu_count == 256
    assert g.render_minor == 128
    assert g.unique_id == 0xABC001


def test_xgmi_peers(tmp_path):
    topo = KfdTopology(str(build_fixture(tmp_path)))
    g1, g2 = topo.gpu_nodes()
    assert g1.xgmi_peers() == [2]
    assert g2.xgmi_peers() == [1]


def test_bdf_decoding(tmp_path):
    topo = KfdTopology(str(build_fixture(tmp_path)))
    g = topo.gpu_nodes()[0]
    # location_id 0x0400 -> bus 0x04, dev 0, fn 0
    assert g.bdf == "0000:04:00.0"


def test_card_minor_mapping(tmp_path):
    topo = KfdTopology(str(build_fixture(tmp_path)))
    assert topo.card_minor_for_render(128) == 0
    assert topo.card_minor_for_render(129) == 1
    assert topo.card_minor_for_render(999) == -1


def test_missing_tree_is_empty(tmp_path):
    topo = KfdTopology(str(tmp_path / "nope"))
    assert not topo.available()
    assert topo.nodes() == []


class TestKfdOnlyBackend:
    """Degraded-mode HAL over the fixture tree (no libamd_smi)."""

    def _fixture_with_vram(self, root):
        build_fixture(root)
        nodes = os.path.join(root, "class", "kfd", "kfd", "topology", "nodes")
        for i in (1, 2):
            banks = os.path.join(nodes, str(i), "mem_banks", "0")
            os.makedirs(banks)
            with open(os.path.join(banks, "properties"), "w") as f:
                f.write(f"heap_type 1\nsize_in_bytes {288 * 1024**3}\n")
        return root

    def test_enumerate(self, tmp_path):
        from k8s_dra_driver_amd.hal.kfd import KfdDeviceLib

        lib = KfdDeviceLib(str(self._fixture_with_vram(tmp_path)))
        lib.open()
        gpus = lib.enumerate()
        assert len(gpus) == 2
        g = gpus[0]
        assert g.architecture == "gfx950"
        assert g.vram_total_mib == 288 * 1024
        assert g.cu_count == 256
        assert g.render_minor == 128
        assert g.uuid.startswith("kfd-")
        assert gpus[0].xgmi_peer_oam_ids() == [1]

    def test_partition_control_refused(self, tmp_path):
        from k8s_dra_driver_amd.hal.base import HalNotSupported
        from k8s_dra_driver_amd.hal.kfd import KfdDeviceLib

        lib = KfdDeviceLib(str(self._fixture_with_vram(tmp_path)))
        lib.open()
        import pytest as _pytest

        with _pytest.raises(HalNotSupported):
            lib.set_compute_partition(0, "CPX")

    def test_unavailable_without_kfd(self, tmp_path):
        from k8s_dra_driver_amd.hal.base import HalUnavailable
        from k8s_dra_driver_amd.hal.kfd import KfdDeviceLib

        import pytest as _pytest

        with _pytest.raises(HalUnavailable):
            KfdDeviceLib(str(tmp_path / "empty")).open()

    def test_full_driver_over_kfd_backend(self, tmp_path):
        """The whole prepare path runs on the degraded backend."""
        from k8s_dra_driver_amd.hal.kfd import KfdDeviceLib
        from k8s_dra_driver_amd.kube.client import InMemoryKube
        from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver

        lib = KfdDeviceLib(str(self._fixture_with_vram(tmp_path / "sys")))
        lib.open()
        kube = InMemoryKube()
        driver = Driver(
            lib,
            kube,
            node_name="n",
            cdi_root=str(tmp_path / "cdi"),
            checkpoint_root=str(tmp_path / "state"),
            use_tmpfs=False,
        )
        driver.startup()
        devs = kube.list_resource_slices("gpu.amd.com")[0]["spec"]["devices"]
        assert len(devs) == 2
        kube.put_resource_claim(
            {
                "metadata": {"namespace": "d", "name": "c", "uid": "u"},
                "status": {
                    "allocation": {
                        "devices": {
                            "results": [
                                {
                                    "request": "g",
                                    "driver": "gpu.amd.com",
                                    "pool": "n",
                                    "device": devs[0]["name"],
                                }
                            ]
                        }
                    }
                },
            }
        )
        res = driver.node_prepare_resources([ClaimRef("d", "c", "u")])["u"]
        assert not res.error
