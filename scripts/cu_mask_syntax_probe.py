#!/usr/bin/env python3
"""Probe which HSA_CU_MASK syntaxes ROCR honors on this box.

The SharedCompute supervisor emits `<gpu>:<0xHEX>` entries (proven to
quarter MFMA throughput on gfx950 in round 1). Multi-GPU sessions need a
single env var, so this probe measures whether `;`-joined entries and
decimal-range syntax also constrain CUs. Run on a GPU box:
    python scripts/cu_mask_syntax_probe.py
Writes JSON to stdout: {syntax: tflops}.
"""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

QUARTER_HEX = "0x" + "f" * 16 + "0" * 48  # CUs 192..255? no — see below
# CUs 0..63 of 256: low 64 bits set, as a 256-bit hex literal
QUARTER_HEX = "0x" + "0" * 48 + "f" * 16

CASES = {
    "none": None,
    "hex_single": f"0:{QUARTER_HEX}",
    "hex_joined_multi_gpu": f"0:{QUARTER_HEX};1:{QUARTER_HEX}",
    "range_single": "0:0-63",
}


def tflops(mask):
    code = (
        f"import sys; sys.path.insert(0, {REPO!r});"
        "import torch;"
        "from k8s_dra_driver_amd import _hiphealth;"
        "print(_hiphealth.mfma_tflops(0, 1024, 2048))"
    )
    env = dict(os.environ)
    env.pop("HSA_CU_MASK", None)
    if mask is not None:
        env["HSA_CU_MASK"] = mask
    out = subprocess.run(
        [sys.executable, "-c", code],
        capture_output=True,
        text=True,
        env=env,
        timeout=300,
    )
    if out.returncode != 0:
        return {"error": out.stderr[-500:]}
    return float(out.stdout.strip().splitlines()[-1])


def main():
    results = {name: tflops(mask) for name, mask in CASES.items()}
    print(json.dumps(results, indent=2))


if __name__ == "__main__":
    main()
