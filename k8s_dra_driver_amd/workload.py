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


def _allreduce_worker(rank, world, nelem, results):
    """One process per visible GPU; RCCL ring all-reduce over xGMI."""
    import os as _os
    import time

    import torch
    import torch.distributed as dist

    _os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    _os.environ.setdefault("MASTER_PORT", "29781")
    dist.init_process_group("nccl", rank=rank, world_size=world)
    torch.cuda.set_device(rank)
    x = torch.ones(nelem, dtype=torch.bfloat16, device=f"cuda:{rank}")
    for _ in range(3):  # warmup
        dist.all_reduce(x)
    torch.cuda.synchronize()
    iters = 20
    t0 = time.perf_counter()
    for _ in range(iters):
        dist.all_reduce(x)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # ring all-reduce moves 2*(n-1)/n of the buffer per GPU per iteration
    bytes_moved = 2 * (world - 1) / world * nelem * 2 * iters
    # each all-reduce multiplies by world; bf16 rounds non-power-of-two
    # worlds, so compare with tolerance
    expected = float(world) ** (iters + 3)
    ok = bool(
        torch.allclose(
            x.float(),
            torch.full_like(x, expected, dtype=torch.float32),
            rtol=0.05,
        )
    )
    results[rank] = (bytes_moved / dt / 1e9, ok)
    dist.destroy_process_group()


def run_allreduce(size_mb: int) -> int:
    """RCCL all-reduce across every visible GPU (the xGMI fabric demo for
    multi-GPU topology claims, gpu-test7)."""
    import torch
    import torch.multiprocessing as mp

    world = torch.cuda.device_count()
    if world == 0:
        print("ERROR: no HIP devices", file=sys.stderr)
        return 1
    nelem = size_mb * 1024 * 1024 // 2  # bf16
    if world == 1:
        print("1 visible GPU: all-reduce is a no-op; device healthy")
        return 0
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(
            _allreduce_worker,
            args=(world, nelem, results),
            nprocs=world,
            join=True,
        )
        per_gpu = dict(results)
    for r, (gbps, ok) in sorted(per_gpu.items()):
        print(f"rank {r}: busbw {gbps:.1f} GB/s numerics_ok={ok}")
    return 0 if all(ok for _, ok in per_gpu.values()) else 1


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("amd-dra-workload")
    ap.add_argument("--list", action="store_true", help="print visible devices (nvidia-smi -L analog)")
    ap.add_argument("--benchmark", action="store_true", help="run bandwidth + MFMA probes")
    ap.add_argument("--burn-ms", type=int, default=0, help="occupancy burn duration")
    ap.add_argument(
        "--nbody",
        type=int,
        default=0,
        help="run the all-pairs n-body benchmark with this many bodies "
        "(the reference demo's nbody --benchmark analog)",
    )
    ap.add_argument(
        "--allreduce-mb",
        type=int,
        default=0,
        help="RCCL all-reduce of this buffer size across visible GPUs",
    )
    args = ap.parse_args(argv)

    info = visible_devices()
    print(json.dumps(info))
    if not info["kfd"] or not info["render_nodes"]:
        print("ERROR: no GPU devices injected", file=sys.stderr)
        return 1
    if args.benchmark or args.burn_ms or args.nbody:
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
            if args.nbody:
                r = _hiphealth.nbody_benchmark(d, args.nbody, 10)
                print(
                    f"device {d}: nbody {r['bodies']} bodies x {r['iters']} "
                    f"iters = {r['gflops']:.0f} GFLOP/s finite={r['finite']}"
                )
    if args.allreduce_mb:
        return run_allreduce(args.allreduce_mb)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
