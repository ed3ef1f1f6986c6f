"""Multi-GPU readiness: the exact launch shape the driver
uses for the round-end SCALE run, exercised on CPU with the gloo backend
at world sizes 2/4/8, so an 8-GPU node needs zero new code.

The driver launches:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N
        --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...
"""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_bench_launch(nproc: int, extra=(), timeout=280):
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={nproc}",
        "--master-addr",
        "127.0.0.1",
        "--master-port",
        str(_free_port()),
        "bench.py",
        "--gpus",
        str(nproc),
        "--steps",
        "2",
        "--warmup",
        "1",
        "--pods-per-step",
        "4",
        *extra,
    ]
    out = subprocess.run(
        cmd, capture_output=True, text=True, cwd=REPO, timeout=timeout
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected:\n{out.stdout}"
    return json.loads(lines[0])


@pytest.mark.timeout(280)
@pytest.mark.parametrize("nproc", [2, 4])
def test_bench_launch_shape_gloo(nproc):
    d = _run_bench_launch(nproc)
    assert d["n_gpus"] == nproc
    assert d["value"] > 0
    assert d["scaling"] == "weak"
    # whole-job aggregate: pods_total covers every rank
    assert d["config"]["global_batch"] == nproc * 4 * 2


@pytest.mark.timeout(280)
def test_bench_launch_shape_gloo_8rank():
    """The full 8-GPU shape of SCALE_rNN (one rank per GPU on one node)."""
    d = _run_bench_launch(8)
    assert d["n_gpus"] == 8
    assert d["value"] > 0


@pytest.mark.timeout(280)
def test_bench_launch_shape_topo4_distributed():
    """Config #5 under the distributed launch (each rank allocates a
    4-adjacent subset through CEL)."""
    d = _run_bench_launch(2, extra=("--config", "topo4"))
    assert d["n_gpus"] == 2
    assert d["config"]["model"] == "dra-claim-lifecycle/topo4"
    assert d["value"] > 0


@pytest.mark.timeout(280)
def test_bench_launch_shape_autocpx_distributed():
    """Scheduler-driven carve mode under the distributed launch: each
    rank auto-carves its own GPU, drains, restores."""
    d = _run_bench_launch(2, extra=("--config", "autocpx", "--hal", "fake"))
    assert d["n_gpus"] == 2
    assert d["config"]["model"] == "dra-claim-lifecycle/autocpx"
    assert d["value"] > 0
