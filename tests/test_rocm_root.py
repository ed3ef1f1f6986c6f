"""ROCm driver-root discovery (reference root.go:29-98 analog)."""

import os

import pytest

from k8s_dra_driver_amd.cdi.rocmroot import discover_rocm_root


def _mk_rocm(root, name, with_runtime=True, version_file=""):
    d = root / name
    (d / "lib").mkdir(parents=True)
    if with_runtime:
        (d / "lib" / "libamdhip64.so.7").write_bytes(b"")
    if version_file:
        (d / ".info").mkdir()
        (d / ".info" / "version").write_text(version_file)
    return d


def test_explicit_path(tmp_path):
    d = _mk_rocm(tmp_path, "myrocm", version_file="7.2.0")
    path, ver = discover_rocm_root(str(d), env={})
    assert path == str(d) and ver == "7.2.0"


def test_explicit_path_without_runtime_is_rejected(tmp_path):
    d = _mk_rocm(tmp_path, "empty", with_runtime=False)
    assert discover_rocm_root(str(d), env={}) is None


def test_env_rocm_path_wins(tmp_path):
    d = _mk_rocm(tmp_path, "env-rocm")
    _mk_rocm(tmp_path / "opt", "rocm")
    found = discover_rocm_root(
        "auto", host_root=str(tmp_path), env={"ROCM_PATH": str(d)}
    )
    assert found[0] == str(d)


def test_opt_rocm_symlink(tmp_path):
    real = _mk_rocm(tmp_path / "opt", "rocm-7.2.0")
    (tmp_path / "opt" / "rocm").symlink_to(real)
    path, ver = discover_rocm_root("auto", host_root=str(tmp_path), env={})
    assert path == f"{tmp_path}/opt/rocm"
    assert ver == "7.2.0"  # from the resolved dir name


def test_highest_version_picked(tmp_path):
    _mk_rocm(tmp_path / "opt", "rocm-6.4.1")
    _mk_rocm(tmp_path / "opt", "rocm-7.10.0")
    _mk_rocm(tmp_path / "opt", "rocm-7.2.0")
    path, _ = discover_rocm_root("auto", host_root=str(tmp_path), env={})
    assert path.endswith("rocm-7.10.0")


def test_nothing_found(tmp_path):
    assert discover_rocm_root("auto", host_root=str(tmp_path), env={}) is None


def test_real_rocm_if_present():
    """On images with /opt/rocm (this one), discovery must find it."""
    if not os.path.isdir("/opt/rocm/lib"):
        pytest.skip("no ROCm on this machine")
    found = discover_rocm_root("auto", env={})
    assert found is not None
    assert found[0].startswith("/opt/rocm")


def test_driver_wires_discovered_mount(tmp_path):
    from k8s_dra_driver_amd.cdi.spec import read_spec_file
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    d = _mk_rocm(tmp_path, "hostrocm", version_file="7.2.0")
    lib = FakeDeviceLib()
    lib.open()
    driver = Driver(
        lib,
        InMemoryKube(),
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        rocm_mount=str(d),
    )
    driver.startup()
    spec = read_spec_file(str(tmp_path / "cdi" / "k8s.gpu.amd.com-device.json"))
    mounts = spec["containerEdits"]["mounts"]
    assert any(
        m["hostPath"] == str(d) and m["containerPath"] == "/opt/rocm"
        for m in mounts
    )
    assert "ROCM_PATH=/opt/rocm" in spec["containerEdits"]["env"]
    driver.shutdown(unpublish=False)


def test_driver_rejects_bad_explicit_mount(tmp_path):
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    lib = FakeDeviceLib()
    lib.open()
    with pytest.raises(RuntimeError, match="no ROCm userspace"):
        Driver(
            lib,
            InMemoryKube(),
            node_name="n",
            cdi_root=str(tmp_path / "cdi"),
            checkpoint_root=str(tmp_path / "state"),
            use_tmpfs=False,
            rocm_mount=str(tmp_path / "nonexistent"),
        )
