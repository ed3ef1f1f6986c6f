"""Demo workload CLI — the nbody-sample analog.

Runs inside claim-bearing pods: prints which devices the CDI injection made
visible, then optionally burns compute / measures bandwidth on them, so the
demo specs can validate sharing and partitioning exactly like the
reference's pods validate with ``nvidia-smi -L`` + the nbody benchmark
(reference gpu-test1.yaml:33-37, gpu-test5.yaml:53-87).
"""

from __future__ import annotations

import argparse
import glob
import json
import os
import sys


def visible_devices() -> dict:
    return {
        "kfd": os.path.exists("/dev/kfd"),
        "render_nodes": sorted(glob.glob("/dev/dri/renderD*")),
        "card_nodes": sorted(glob.glob("/dev/dri/card*")),
        "claim_uid": os.environ.get("AMD_DRA_CLAIM_UID", ""),
        "shared_session": os.environ.get("AMD_DRA_SHARED_SESSION", ""),
        "cu_mask": os.environ.get("HSA_CU_MASK", ""),
    }


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("amd-dra-workload")
    ap.add_argument("--list", action="store_true", help="print visible devices (nvidia-smi -L analog)")
    ap.add_argument("--benchmark", action="store_true", help="run bandwidth + MFMA probes")
    ap.add_argument("--burn-ms", type=int, default=0, help="occupancy burn duration")
    args = ap.parse_args(argv)

    info = visible_devices()
    print(json.dumps(info))
    if not info["kfd"] or not info["render_nodes"]:
        print("ERROR: no GPU devices injected", file=sys.stderr)
        return 1
    if args.benchmark or args.burn_ms:
        from k8s_dra_driver_amd import _hiphealth

        n = _hiphealth.device_count()
        print(f"hip devices: {n}")
        for d in range(n):
            if args.benchmark:
                bw = _hiphealth.bandwidth_gbs(d, 256, 5)
                chk = _hiphealth.mfma_check(d)
                print(f"device {d}: bandwidth {bw:.0f} GB/s mfma_ok={chk['ok']}")
            if args.burn_ms:
                ms = _hiphealth.burn_ms(d, args.burn_ms)
                print(f"device {d}: burned {ms:.0f} ms")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
