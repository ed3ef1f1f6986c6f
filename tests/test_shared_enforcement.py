"""Shared-GPU isolation enforcement: the supervisor-side
detector that catches containers stripping or altering their CU mask —
the out-of-band loop the reference gets from its MPS daemon
(sharing.go:211-221)."""

import os

import pytest

from k8s_dra_driver_amd.api.types import SharedComputeSettings
from k8s_dra_driver_amd.hal import FakeDeviceLib
from k8s_dra_driver_amd.hal.model import AllocatableDevice
from k8s_dra_driver_amd.sharing.enforce import SharedEnforcer, Violation
from k8s_dra_driver_amd.sharing.shared import SharedComputeManager


def _gpu_devices(lib, n=1):
    return [AllocatableDevice.from_gpu(g) for g in lib.enumerate()[:n]]


@pytest.fixture
def setup(tmp_path):
    lib = FakeDeviceLib()
    lib.open()
    mgr = SharedComputeManager(root=str(tmp_path / "shared"), use_tmpfs=False)
    session = mgr.start_session(
        "claim-abc", _gpu_devices(lib), SharedComputeSettings(default_cu_share_percent=25)
    )
    proc = tmp_path / "proc"
    kfd = tmp_path / "kfd-proc"
    kfd.mkdir()
    return lib, mgr, session, proc, kfd


def _fake_pid(proc, kfd, pid, env: dict):
    (kfd / str(pid)).mkdir()
    d = proc / str(pid)
    d.mkdir(parents=True)
    (d / "environ").write_bytes(
        b"\0".join(f"{k}={v}".encode() for k, v in env.items()) + b"\0"
    )


def _enforcer(mgr, proc, kfd, **kw):
    return SharedEnforcer(
        mgr, proc_root=str(proc), kfd_proc_root=str(kfd), **kw
    )


def _session_env(session) -> dict:
    return dict(e.split("=", 1) for e in session.env)


def test_compliant_process_passes(setup):
    lib, mgr, session, proc, kfd = setup
    env = _session_env(session)
    env["AMD_DRA_CLAIM_UID"] = "claim-abc"
    _fake_pid(proc, kfd, 100, env)
    assert _enforcer(mgr, proc, kfd).scan() == []


def test_stripped_mask_detected(setup):
    lib, mgr, session, proc, kfd = setup
    env = _session_env(session)
    env.pop("HSA_CU_MASK")  # adversarial container scrubbed its slice
    env["AMD_DRA_CLAIM_UID"] = "claim-abc"
    _fake_pid(proc, kfd, 101, env)
    seen = []
    enf = _enforcer(mgr, proc, kfd, on_violation=seen.append)
    violations = enf.scan()
    assert [v.kind for v in violations] == ["stripped"]
    assert violations[0].pid == 101
    assert violations[0].claim_uid == "claim-abc"
    assert seen == violations
    assert enf.violation_count == 1


def test_altered_mask_detected(setup):
    lib, mgr, session, proc, kfd = setup
    env = _session_env(session)
    env["HSA_CU_MASK"] = "0:0xffffffffffffffff"  # grabbed a bigger slice
    _fake_pid(proc, kfd, 102, env)
    violations = _enforcer(mgr, proc, kfd).scan()
    assert [v.kind for v in violations] == ["altered"]


def test_orphan_session_detected(setup):
    lib, mgr, session, proc, kfd = setup
    _fake_pid(
        proc,
        kfd,
        103,
        {"AMD_DRA_SHARED_SESSION": "no-such-session", "HSA_CU_MASK": "0:0x1"},
    )
    violations = _enforcer(mgr, proc, kfd).scan()
    assert [v.kind for v in violations] == ["orphan-session"]


def test_non_shared_gpu_process_ignored(setup):
    """Whole-GPU claims and host processes carry no session marker and are
    not the enforcer's concern."""
    lib, mgr, session, proc, kfd = setup
    _fake_pid(proc, kfd, 104, {"PATH": "/usr/bin"})
    assert _enforcer(mgr, proc, kfd).scan() == []


def test_exited_process_skipped(setup):
    lib, mgr, session, proc, kfd = setup
    (kfd / "105").mkdir()  # in KFD listing but no /proc entry anymore
    assert _enforcer(mgr, proc, kfd).scan() == []


def test_kill_action(setup):
    lib, mgr, session, proc, kfd = setup
    import subprocess
    import sys

    victim = subprocess.Popen([sys.executable, "-c", "import time; time.sleep(60)"])
    try:
        env = _session_env(session)
        env.pop("HSA_CU_MASK")
        _fake_pid(proc, kfd, victim.pid, env)
        enf = _enforcer(mgr, proc, kfd, action="kill")
        violations = enf.scan()
        assert [v.kind for v in violations] == ["stripped"]
        assert victim.wait(timeout=10) == -9  # SIGKILLed
    finally:
        if victim.poll() is None:
            victim.kill()


def test_no_kfd_directory_is_noop(tmp_path):
    mgr = SharedComputeManager(root=str(tmp_path / "s"), use_tmpfs=False)
    enf = SharedEnforcer(
        mgr,
        proc_root=str(tmp_path / "proc"),
        kfd_proc_root=str(tmp_path / "absent"),
    )
    assert enf.scan() == []


def test_driver_emits_warning_event(tmp_path):
    """Driver wiring: a violation becomes a Warning event on the claim +
    a metric increment."""
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "state"),
        use_tmpfs=False,
        shared_enforcement="warn",
    )
    assert driver.enforcer is not None
    driver._on_isolation_violation(
        Violation(
            pid=4242,
            kind="stripped",
            session_id="s",
            claim_uid="uid-x",
            detail="HSA_CU_MASK stripped",
        )
    )
    assert len(kube.events) == 1
    ev = kube.events[0]
    assert ev["type"] == "Warning"
    assert ev["reason"] == "SharedIsolationViolation"
    assert "4242" in ev["message"]
    driver.shutdown(unpublish=False)


def test_gpu_indices_scoping(tmp_path):
    """nvkind analog: two plugin instances on one box, each publishing a
    disjoint GPU subset as its own 'node' (reference values.yaml:40-48
    maskNvidiaDriverParams trick)."""
    from k8s_dra_driver_amd import DRIVER_NAME
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import Driver

    kube = InMemoryKube()
    drivers = []
    for node, idxs in (("node-a", [0, 1, 2, 3]), ("node-b", [4, 5, 6, 7])):
        lib = FakeDeviceLib()
        lib.open()
        d = Driver(
            lib,
            kube,
            node_name=node,
            cdi_root=str(tmp_path / node / "cdi"),
            checkpoint_root=str(tmp_path / node / "state"),
            use_tmpfs=False,
            gpu_indices=idxs,
        )
        d.startup()
        drivers.append(d)
    slices = kube.list_resource_slices(DRIVER_NAME)
    by_node = {}
    for s in slices:
        names = [d["name"] for d in s["spec"]["devices"]]
        by_node.setdefault(s["spec"]["nodeName"], []).extend(names)
    assert sorted(by_node) == ["node-a", "node-b"]
    assert by_node["node-a"] == ["gpu-0", "gpu-1", "gpu-2", "gpu-3"]
    assert by_node["node-b"] == ["gpu-4", "gpu-5", "gpu-6", "gpu-7"]
    for d in drivers:
        d.shutdown(unpublish=False)


def test_enforcer_background_loop(setup):
    """start() scans periodically; violations surface without manual
    scan() calls (the daemon-loop MPS parity)."""
    import time

    lib, mgr, session, proc, kfd = setup
    env = _session_env(session)
    env.pop("HSA_CU_MASK")
    _fake_pid(proc, kfd, 201, env)
    seen = []
    enf = _enforcer(
        mgr, proc, kfd, interval_s=0.05, on_violation=seen.append
    )
    enf.start()
    try:
        t0 = time.time()
        while not seen and time.time() - t0 < 5:
            time.sleep(0.02)
        assert seen and seen[0].pid == 201
    finally:
        enf.stop()
    assert enf._thread is None
