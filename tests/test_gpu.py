"""GPU-marked tests: run on a real MI355X via gpurun.

These exercise the NATIVE path end-to-end: the _amdhal amdsmi binding, KFD
sysfs topology, real device nodes in CDI specs, and the gfx950 HIP health
kernels. They fail loudly (no eager/fake fallback) if the native
extensions are missing.
"""

import os

import pytest

# torch must load before the system-ROCm-linked _hiphealth extension: its
# bundled HIP runtime wins the dlopen race, otherwise torch.cuda reports
# "No HIP GPUs are available" (observed on the MI355X pool, round 1).
import torch  # noqa: F401  (import order is load-bearing)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def real_lib():
    from k8s_dra_driver_amd.hal.amdsmi import AmdSmiDeviceLib

    lib = AmdSmiDeviceLib()
    lib.open()
    yield lib
    lib.close()


class TestNativeEnumeration:
    def test_amdhal_binds(self):
        from k8s_dra_driver_amd import _amdhal

        v = _amdhal.lib_version()
        assert v["major"] >= 24

    def test_enumerate_real_gpu(self, real_lib):
        gpus = real_lib.enumerate()
        assert len(gpus) >= 1
        g = gpus[0]
        assert g.architecture == "gfx950"
        assert g.vram_total_mib > 200 * 1024  # 288 GiB HBM3E
        assert g.cu_count == 256
        assert g.render_minor >= 128
        assert g.uuid
        assert g.compute_partition in ("SPX", "DPX", "TPX", "QPX", "CPX")
        assert g.memory_partition.startswith("NPS")

    def test_kfd_topology_matches_amdsmi(self, real_lib):
        nodes = real_lib.topology.gpu_nodes()
        assert len(nodes) >= 1
        assert nodes[0].gfx_arch == "gfx950"
        minors = {n.render_minor for n in nodes}
        for g in real_lib.enumerate():
            assert g.render_minor in minors

    def test_device_nodes_exist(self, real_lib):
        paths = real_lib.device_node_paths(0)
        assert os.path.exists(paths["kfd"])
        assert os.path.exists(paths["renderD"])

    def test_health_check(self, real_lib):
        h = real_lib.health_check(0)
        assert h["status"] == "healthy"


class TestPreparePathOnHardware:
    def test_full_prepare_with_real_hal(self, real_lib, tmp_path):
        from k8s_dra_driver_amd.cdi.spec import read_spec_file
        from k8s_dra_driver_amd.kube.client import InMemoryKube
        from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver

        kube = InMemoryKube()
        driver = Driver(
            real_lib,
            kube,
            node_name="gpu-node",
            cdi_root=str(tmp_path / "cdi"),
            checkpoint_root=str(tmp_path / "state"),
            use_tmpfs=False,
        )
        driver.startup()
        slices = kube.list_resource_slices("gpu.amd.com")
        assert slices and slices[0]["spec"]["devices"]
        dev_name = slices[0]["spec"]["devices"][0]["name"]

        kube.put_resource_claim(
            {
                "metadata": {
                    "namespace": "default",
                    "name": "hw-claim",
                    "uid": "hw-uid",
                },
                "status": {
                    "allocation": {
                        "devices": {
                            "results": [
                                {
                                    "request": "gpu",
                                    "driver": "gpu.amd.com",
                                    "pool": "gpu-node",
                                    "device": dev_name,
                                }
                            ]
                        }
                    }
                },
            }
        )
        res = driver.node_prepare_resources(
            [ClaimRef("default", "hw-claim", "hw-uid")]
        )["hw-uid"]
        assert not res.error
        base = read_spec_file(str(tmp_path / "cdi" / "k8s.gpu.amd.com-device.json"))
        node_paths = [
            n["path"]
            for d in base["devices"]
            for n in d["containerEdits"]["deviceNodes"]
        ]
        for p in node_paths:
            assert os.path.exists(p), f"CDI references missing node {p}"
        driver.node_unprepare_resources(
            [ClaimRef("default", "hw-claim", "hw-uid")]
        )


class TestHipHealth:
    def test_device_info(self):
        from k8s_dra_driver_amd import _hiphealth

        assert _hiphealth.device_count() >= 1
        info = _hiphealth.device_info(0)
        assert "gfx950" in info["gcn_arch"]
        assert info["warp_size"] == 64
        assert info["multi_processor_count"] == 256

    def test_mfma_correctness(self):
        from k8s_dra_driver_amd import _hiphealth

        check = _hiphealth.mfma_check(0)
        assert check["ok"], check

    def test_bandwidth_sane(self):
        from k8s_dra_driver_amd import _hiphealth

        bw = _hiphealth.bandwidth_gbs(0, 1024, 10)
        # HBM3E: 8 TB/s peak, ~6.3 achievable; anything under 2 TB/s means
        # the copy never hit the real device.
        assert bw > 2000, f"bandwidth {bw} GB/s too low"

    def test_mfma_throughput_sane(self):
        from k8s_dra_driver_amd import _hiphealth

        tf = _hiphealth.mfma_tflops(0, 2048, 2048)
        # dense bf16 peak ~2.5 PF; a health kernel should clear 0.5 PF
        assert tf > 500, f"MFMA throughput {tf} TF/s too low"


class TestCuMaskEnforcement:
    """HSA_CU_MASK is the SharedCompute supervisor's spatial-isolation
    mechanism (docs/sharing.md); verify it really constrains compute on
    gfx950: 64 of 256 CUs must cut MFMA throughput to ~1/4."""

    def _tflops_with_mask(self, mask_env):
        import json
        import subprocess
        import sys

        code = (
            "import sys, json;"
            f"sys.path.insert(0, {os.getcwd()!r});"
            "import torch;"
            "from k8s_dra_driver_amd import _hiphealth;"
            "print(json.dumps(_hiphealth.mfma_tflops(0, 1024, 2048)))"
        )
        env = dict(os.environ)
        if mask_env:
            env["HSA_CU_MASK"] = mask_env
        out = subprocess.run(
            [sys.executable, "-c", code],
            capture_output=True,
            text=True,
            env=env,
            timeout=300,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        return json.loads(out.stdout.strip().splitlines()[-1])

    def test_cu_mask_quarters_throughput(self):
        from k8s_dra_driver_amd.sharing.shared import cu_mask_hex

        full = self._tflops_with_mask(None)
        quarter_mask = f"0:{cu_mask_hex(0, 64, 256)}"
        quarter = self._tflops_with_mask(quarter_mask)
        ratio = quarter / full
        # expect ~0.25; allow clocks/binding slack
        assert 0.15 < ratio < 0.45, (full, quarter, ratio)


class TestTorchInterop:
    def test_torch_sees_gpu(self):
        import torch

        assert torch.cuda.is_available()
        x = torch.randn(256, 256, device="cuda:0", requires_grad=True)
        (x @ x).sum().backward()
        torch.cuda.synchronize()
        assert x.grad is not None


class TestNbodyDemo:
    def test_nbody_benchmark(self):
        """The reference demo's nbody --benchmark analog on gfx950."""
        from k8s_dra_driver_amd import _hiphealth

        r = _hiphealth.nbody_benchmark(0, 65536, 10)
        assert r["finite"], r
        # 256 CUs of fp32 VALU: an LDS-tiled all-pairs kernel should clear
        # 10 TFLOP/s comfortably (peak vector fp32 is ~157 TF)
        assert r["gflops"] > 10_000, r


class TestKfdBackendOnHardware:
    def test_kfd_only_enumeration_matches_amdsmi(self, real_lib):
        """Degraded backend agrees with amdsmi on the basics."""
        from k8s_dra_driver_amd.hal.kfd import KfdDeviceLib

        kfd = KfdDeviceLib()
        kfd.open()
        kfd_gpus = kfd.enumerate()
        smi_gpus = real_lib.enumerate()
        assert len(kfd_gpus) == len(smi_gpus)
        k, s = kfd_gpus[0], smi_gpus[0]
        assert k.architecture == s.architecture == "gfx950"
        assert k.render_minor == s.render_minor
        assert k.cu_count == s.cu_count == 256
        # VRAM from mem_banks within 2% of amdsmi's number
        assert abs(k.vram_total_mib - s.vram_total_mib) < s.vram_total_mib * 0.02


class TestRcclFabric:
    """RCCL-over-xGMI readiness: the allreduce demo the
    multi-GPU topology claims run, wired as a gpu test that auto-skips
    below world-size 2 (the pool's boxes expose 1 GPU; an 8-GPU node
    exercises the full ring)."""

    def test_rccl_allreduce_all_visible_gpus(self):
        if torch.cuda.device_count() < 2:
            pytest.skip("needs >=2 visible GPUs for a ring all-reduce")
        from k8s_dra_driver_amd.workload import run_allreduce

        assert run_allreduce(64) == 0

    def test_timeslice_knob_probe_runs(self, real_lib):
        """The quantum-knob probe must never raise on real hardware, and
        the published attribute must reflect it."""
        eff = real_lib.timeslice_effective()
        assert isinstance(eff, bool)
        g0 = real_lib.enumerate()[0]
        assert g0.timeslice_effective == eff

    def test_repartition_capability_probe_runs(self, real_lib):
        cap = real_lib.dynamic_repartition_capable()
        assert isinstance(cap, bool)
        g0 = real_lib.enumerate()[0]
        assert g0.repartition_capable == cap


class TestSharedEnforcementOnHardware:
    """Adversarial case on real KFD: a process attached to
    the GPU whose env claims a SharedCompute session but carries no CU
    mask must be detected via /sys/class/kfd/kfd/proc attribution."""

    def test_stripped_mask_process_detected(self, real_lib, tmp_path):
        import subprocess
        import sys
        import time as _time

        from k8s_dra_driver_amd.api.types import SharedComputeSettings
        from k8s_dra_driver_amd.hal.model import AllocatableDevice
        from k8s_dra_driver_amd.sharing.enforce import SharedEnforcer
        from k8s_dra_driver_amd.sharing.shared import SharedComputeManager

        if True:
            mgr = SharedComputeManager(
                root=str(tmp_path / "shared"), use_tmpfs=False
            )
            session = mgr.start_session(
                "hw-claim-1",
                [AllocatableDevice.from_gpu(real_lib.enumerate()[0])],
                SharedComputeSettings(default_cu_share_percent=25),
            )
            # adversarial pod process: opens the GPU, claims the session,
            # but scrubbed HSA_CU_MASK from its environment
            env = dict(os.environ)
            env.pop("HSA_CU_MASK", None)
            env["AMD_DRA_SHARED_SESSION"] = session.session_id
            env["AMD_DRA_CLAIM_UID"] = "hw-claim-1"
            code = (
                "import torch, time;"
                "torch.zeros(4, device='cuda:0');"
                "print('attached', flush=True);"
                "time.sleep(120)"
            )
            proc = subprocess.Popen(
                [sys.executable, "-c", code],
                env=env,
                stdout=subprocess.PIPE,
                text=True,
            )
            try:
                assert proc.stdout.readline().strip() == "attached"
                _time.sleep(1.0)  # KFD proc entry settles
                enf = SharedEnforcer(mgr)
                pids = enf.gpu_pids()
                if proc.pid not in pids:
                    # this test environment lacks the host PID namespace:
                    # KFD lists host pids we cannot see in /proc. The
                    # enforcer must detect and surface exactly that
                    # (production runs with hostPID: true via the chart).
                    enf.scan()
                    assert enf.last_unattributable, (
                        f"pid {proc.pid} absent from KFD list {pids} but "
                        "no unattributable pids flagged either"
                    )
                    pytest.skip(
                        "no host PID namespace on this box (container "
                        "pool); attribution verified as unattributable, "
                        "full adversarial check needs hostPID:true"
                    )
                violations = enf.scan()
                mine = [v for v in violations if v.pid == proc.pid]
                assert len(mine) == 1 and mine[0].kind == "stripped", violations
            finally:
                proc.kill()
                proc.wait()

    def test_compliant_process_not_flagged(self, real_lib, tmp_path):
        import subprocess
        import sys
        import time as _time

        from k8s_dra_driver_amd.api.types import SharedComputeSettings
        from k8s_dra_driver_amd.hal.model import AllocatableDevice
        from k8s_dra_driver_amd.sharing.enforce import SharedEnforcer
        from k8s_dra_driver_amd.sharing.shared import SharedComputeManager

        if True:
            mgr = SharedComputeManager(
                root=str(tmp_path / "shared"), use_tmpfs=False
            )
            session = mgr.start_session(
                "hw-claim-2",
                [AllocatableDevice.from_gpu(real_lib.enumerate()[0])],
                SharedComputeSettings(default_cu_share_percent=25),
            )
            env = dict(os.environ)
            for e in session.env:
                k, _, v = e.partition("=")
                env[k] = v
            env["AMD_DRA_CLAIM_UID"] = "hw-claim-2"
            code = (
                "import torch, time;"
                "torch.zeros(4, device='cuda:0');"
                "print('attached', flush=True);"
                "time.sleep(120)"
            )
            proc = subprocess.Popen(
                [sys.executable, "-c", code],
                env=env,
                stdout=subprocess.PIPE,
                text=True,
            )
            try:
                assert proc.stdout.readline().strip() == "attached"
                _time.sleep(1.0)
                violations = SharedEnforcer(mgr).scan()
                assert [v for v in violations if v.pid == proc.pid] == []
            finally:
                proc.kill()
                proc.wait()


class TestLiveCapsEndToEnd:
    """the pool's firmware reports restricted NPS caps
    (round 1 observed NPS1/NPS2 only, not the bare-metal NPS4 default);
    the enumerate() -> catalog path must honor the LIVE caps end to end,
    and the repartition capability probe must agree with what set()
    actually does."""

    def test_published_profiles_respect_live_nps_caps(self, real_lib):
        from k8s_dra_driver_amd.partition.catalog import make_profile

        for g in real_lib.enumerate():
            assert g.nps_caps, "live NPS caps must be reported"
            # any profile the driver would build for this GPU must use a
            # mode the hardware actually reports
            for mem_mode in g.nps_caps:
                prof = make_profile(
                    "CPX" if "CPX" in g.compute_caps else g.compute_partition,
                    mem_mode,
                    vram_total_mib=g.vram_total_mib or 288 * 1024,
                    cu_count=g.cu_count or 256,
                    nps_caps=g.nps_caps,
                )
                assert prof.memory_mode in g.nps_caps
            # modes NOT in the live caps are refused by the catalog
            missing = [m for m in ("NPS1", "NPS2", "NPS4") if m not in g.nps_caps]
            for m in missing:
                import pytest as _pytest

                with _pytest.raises(ValueError):
                    make_profile("CPX", m, nps_caps=g.nps_caps)

    def test_repartition_probe_matches_reality(self, real_lib):
        """If the probe says capable, a same-mode set must succeed; if it
        says incapable, the set must fail — either way they must agree
        (the learned-capability contract)."""
        g0 = real_lib.enumerate()[0]
        probed = real_lib.dynamic_repartition_capable()
        try:
            real_lib.set_compute_partition(0, g0.compute_partition)
            actually = True
        except Exception:
            actually = False
        assert real_lib.dynamic_repartition_capable() == actually
        if probed != actually:
            # probe was optimistic/pessimistic; the learned value must win
            assert real_lib.dynamic_repartition_capable() == actually


class TestNbodyCorrectness:
    """The demo workload's integration cross-checked against a CPU fp32
    reference of the same deterministic init (round-1 weak finding:
    finiteness alone is a thin correctness artifact)."""

    def test_nbody_matches_cpu_reference(self):
        import numpy as np

        from k8s_dra_driver_amd import _hiphealth

        n, iters, sample = 512, 3, 64
        gpu = np.array(
            _hiphealth.nbody_positions(0, n, iters, sample), dtype=np.float32
        )

        # replicate the kernel's LCG init exactly
        s = np.uint32(0x5A1AD)
        vals = np.empty(n * 3, dtype=np.float32)
        sv = int(s)
        for i in range(n * 3):
            sv = (sv * 1664525 + 1013904223) & 0xFFFFFFFF
            vals[i] = np.float32(sv >> 8) / np.float32(1 << 24) - np.float32(
                0.5
            )
        pos = vals.reshape(n, 3).astype(np.float32)
        mass = np.full(n, 1.0 / n, dtype=np.float32)
        vel = np.zeros_like(pos)
        dt, soft2 = np.float32(1e-3), np.float32(1e-4)
        for _ in range(iters):
            d = pos[None, :, :] - pos[:, None, :]  # i -> j
            r2 = (d * d).sum(-1) + soft2
            f = mass[None, :] / (r2 * np.sqrt(r2))
            acc = (f[:, :, None] * d).sum(1)
            vel = vel + acc.astype(np.float32) * dt
            pos = pos + vel * dt
        ref = pos[:sample]
        err = np.abs(gpu - ref).max()
        scale = np.abs(ref).max()
        assert err / scale < 1e-3, (err, scale)


class TestProspectivePartitionsOnHardware:
    """Prospective-partition publication respects the real box's
    repartition capability: pool slices (incapable) must publish no
    prospective devices; bare metal publishes 8 per GPU."""

    def test_gating_matches_capability(self, real_lib, tmp_path):
        from k8s_dra_driver_amd import DRIVER_NAME
        from k8s_dra_driver_amd.kube.client import InMemoryKube
        from k8s_dra_driver_amd.plugin.driver import Driver

        kube = InMemoryKube()
        kube.api_versions = ["v1beta2", "v1beta1"]
        driver = Driver(
            real_lib,
            kube,
            node_name="hw",
            cdi_root=str(tmp_path / "cdi"),
            checkpoint_root=str(tmp_path / "state"),
            use_tmpfs=False,
            prospective_partitions="cpx",
        )
        driver.startup()
        try:
            devs = [
                d["name"]
                for s in kube.list_resource_slices(DRIVER_NAME)
                for d in s["spec"]["devices"]
            ]
            n_gpus = len(real_lib.enumerate())
            prospective = [d for d in devs if "-cpx-" in d]
            if real_lib.dynamic_repartition_capable():
                assert len(prospective) == 8 * n_gpus
            else:
                assert prospective == [], (
                    "incapable box must not advertise carves it cannot do"
                )
        finally:
            driver.shutdown(unpublish=False)
