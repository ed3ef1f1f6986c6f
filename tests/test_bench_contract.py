"""Guards the bench.py driver contract: flags, JSON schema, single line."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_bench_json_contract():
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--steps",
            "2",
            "--warmup",
            "1",
            "--pods-per-step",
            "4",
        ],
        capture_output=True,
        text=True,
        cwd=REPO,
        timeout=280,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    for key in (
        "metric",
        "value",
        "unit",
        "n_gpus",
        "steps",
        "warmup",
        "ms_per_step",
        "higher_is_better",
        "scaling",
        "vs_baseline",
        "dtype",
        "data",
        "config",
    ):
        assert key in d, key
    assert d["metric"] == "gpu_pods_scheduled_per_sec"
    assert d["n_gpus"] == 1
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None  # reference publishes no number
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    assert cfg["model"] == "dra-claim-lifecycle/whole"
    assert cfg["hal"] in ("fake", "amdsmi")
    assert cfg["alloc_prepare_p50_ms"] > 0


def test_metrics_histograms():
    from k8s_dra_driver_amd.metrics.prom import PluginMetrics
    from prometheus_client import generate_latest

    m = PluginMetrics()
    with m.time_prepare():
        pass
    m.prepared_claims.inc()
    m.allocatable_devices.set(8)
    text = generate_latest(m.registry).decode()
    assert "dra_prepare_seconds_bucket" in text
    assert "dra_prepared_claims_total 1.0" in text
    assert "dra_allocatable_devices 8.0" in text


def test_repartition_metric_counts(tmp_path):
    from k8s_dra_driver_amd.api.types import API_GROUP_VERSION
    from k8s_dra_driver_amd.hal import FakeDeviceLib
    from k8s_dra_driver_amd.kube.client import InMemoryKube
    from k8s_dra_driver_amd.plugin.driver import ClaimRef, Driver
    from prometheus_client import generate_latest

    lib = FakeDeviceLib()
    lib.open()
    kube = InMemoryKube()
    driver = Driver(
        lib,
        kube,
        node_name="n",
        cdi_root=str(tmp_path / "cdi"),
        checkpoint_root=str(tmp_path / "s"),
        use_tmpfs=False,
    )
    driver.startup()
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
                                "device": "gpu-0",
                            }
                        ],
                        "config": [
                            {
                                "source": "FromClaim",
                                "requests": [],
                                "opaque": {
                                    "driver": "gpu.amd.com",
                                    "parameters": {
                                        "apiVersion": API_GROUP_VERSION,
                                        "kind": "PartitionConfig",
                                        "computePartition": "CPX",
                                        "memoryPartition": "NPS1",
                                        "allowDynamicRepartition": True,
                                    },
                                },
                            }
                        ],
                    }
                }
            },
        }
    )
    driver.node_prepare_resources([ClaimRef("d", "c", "u")])
    driver.node_unprepare_resources([ClaimRef("d", "c", "u")])
    text = generate_latest(driver.metrics.registry).decode()
    # one switch at prepare + one restore at unprepare
    assert "dra_repartitions_total 2.0" in text


def test_bench_all_configs_attested():
    """BASELINE configs #1-#5 are first-class bench modes:
    --config all emits the whole-GPU headline plus per-config results."""
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--steps", "2", "--warmup", "1", "--pods-per-step", "8",
            "--config", "all",
        ],
        capture_output=True,
        text=True,
        cwd=REPO,
        timeout=280,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["config"]["model"] == "dra-claim-lifecycle/whole"
    configs = d["config"]["configs"]
    for name in ("mock", "shared", "cpx", "topo4"):
        assert name in configs, name
        sub = configs[name]
        assert "error" not in sub, f"{name}: {sub}"
        assert sub["pods_per_sec"] > 0
        assert sub["alloc_prepare_p50_ms"] > 0
    assert configs["mock"]["hal"] == "fake"


def test_bench_hal_amdsmi_hard_fails_without_gpu():
    """--hal amdsmi must never silently bench the fake backend."""
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--steps", "1", "--warmup", "0", "--pods-per-step", "1",
            "--hal", "amdsmi",
        ],
        capture_output=True,
        text=True,
        cwd=REPO,
        timeout=280,
    )
    if out.returncode == 0:  # only on a real GPU box
        d = json.loads(out.stdout.strip().splitlines()[-1])
        assert d["config"]["hal"] == "amdsmi"
    else:
        assert "amdsmi" in out.stderr
