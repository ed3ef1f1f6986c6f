"""amd-dra-demo: the three-process clusterless demo harness."""

import os

import pytest

from k8s_dra_driver_amd.demo_runner import main

SPECS = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "demo",
    "specs",
    "quickstart",
)


@pytest.mark.timeout(240)
def test_demo_runner_end_to_end(capsys):
    rc = main(
        [
            "--carve",
            "gpu-7:CPX",
            os.path.join(SPECS, "gpu-test1.yaml"),
            os.path.join(SPECS, "gpu-test4.yaml"),
            os.path.join(SPECS, "gpu-test5.yaml"),
            os.path.join(SPECS, "partition-carve.yaml"),
        ]
    )
    assert rc == 0
    out = capsys.readouterr().out
    assert "system up: 8 device(s)" in out
    assert "POD pod1: devices=['gpu-0']" in out
    assert "HSA_CU_MASK" in out  # shared-compute pod env
    assert "cpx-" in out  # dynamic carve + pre-carved partitions
    # gpu-test4: four partitions constrained to ONE parent die
    import re

    m = re.search(r"gpu-test4.*?devices=\[([^\]]+)\]", out, re.S)
    assert m, out
    parents = {d.strip(" '").rsplit("-", 2)[0] for d in m.group(1).split(",")}
    assert len(parents) == 1
