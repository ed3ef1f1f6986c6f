"""amd-dra-ctl — operational CLI for the driver's hardware layer.

The nvidia-smi-adjacent workflows an operator needs around this driver:
inspect devices/topology as the driver sees them, dump the ResourceSlice
projection, carve/restore partitions, and run the health probes.

    amd-dra-ctl list                 # devices incl. partitions
    amd-dra-ctl topology             # xGMI adjacency matrix
    amd-dra-ctl slice                # ResourceSlice device JSON
    amd-dra-ctl partition 0 CPX NPS1 # dynamic repartition (needs idle GPU)
    amd-dra-ctl health               # per-GPU health + HIP probes
Use ``--hal fake`` anywhere to drive the modeled 8xMI355X node.
"""

from __future__ import annotations

import argparse
import json
import sys

from .hal import new_device_lib
from .hal.model import AllocatableDevice


def _lib(args):
    lib = new_device_lib(args.hal)
    lib.open()
    return lib


def cmd_list(args) -> int:
    lib = _lib(args)
    for g in lib.enumerate():
        print(
            f"{g.canonical_name}: {g.product_name} [{g.architecture}] "
            f"uuid={g.uuid} oam={g.oam_id} vram={g.vram_total_mib}MiB "
            f"cu={g.cu_count} renderD{g.render_minor} "
            f"mode={g.compute_partition}/{g.memory_partition}"
        )
        for p in g.partitions:
            print(
                f"  {p.canonical_name}: kfd={p.kfd_node_id} "
                f"renderD{p.render_minor} "
                f"cu={p.profile.cus_per_partition} "
                f"mem={p.profile.memory_mib_per_partition}MiB "
                f"domain={p.profile.memory_domain_of(p.partition_id)}"
            )
    return 0


def cmd_topology(args) -> int:
    lib = _lib(args)
    gpus = lib.enumerate()
    ids = [g.oam_id for g in gpus]
    print("xGMI adjacency (links between OAM ids):")
    print("     " + " ".join(f"{i:>3}" for i in ids))
    for g in gpus:
        peers = set(g.xgmi_peer_oam_ids())
        row = " ".join(
            "  x" if i == g.oam_id else ("  1" if i in peers else "  .")
            for i in ids
        )
        print(f"{g.oam_id:>4} {row}")
    hives = {g.xgmi_hive_id for g in gpus}
    print(f"hives: {sorted(hives)}")
    return 0


def cmd_slice(args) -> int:
    lib = _lib(args)
    devices = []
    shared_counters = []
    for g in lib.enumerate():
        if g.partitions:
            devices.extend(
                AllocatableDevice.from_partition(g, p).to_device()
                for p in g.partitions
            )
        else:
            if getattr(args, "prospective", ""):
                from .hal.model import (
                    gpu_device_with_counters,
                    prospective_partition_devices,
                    shared_counter_set,
                )
                from .partition.catalog import (
                    make_profile,
                    preferred_memory_mode,
                )

                shared_counters.append(shared_counter_set(g))
                devices.append(gpu_device_with_counters(g))
                prof = make_profile(
                    args.prospective.upper(),
                    preferred_memory_mode(
                        args.prospective.upper(), g.nps_caps
                    ),
                    vram_total_mib=g.vram_total_mib or 288 * 1024,
                    cu_count=g.cu_count or 256,
                    nps_caps=g.nps_caps,
                )
                devices.extend(prospective_partition_devices(g, prof))
            else:
                devices.append(AllocatableDevice.from_gpu(g).to_device())
    out: dict = {"devices": devices}
    if shared_counters:
        out["sharedCounters"] = shared_counters
    print(json.dumps(out, indent=2, sort_keys=True))
    return 0


def cmd_partition(args) -> int:
    from .partition.manager import PartitionManager, RepartitionRefused

    lib = _lib(args)
    mgr = PartitionManager(lib)
    try:
        switched = mgr.ensure_mode(
            args.gpu, args.compute.upper(), args.memory.upper(), allow_dynamic=True
        )
    except (RepartitionRefused, ValueError) as e:
        print(f"refused: {e}", file=sys.stderr)
        return 1
    g = lib.enumerate()[args.gpu]
    print(
        f"gpu-{args.gpu}: {'switched to' if switched else 'already'} "
        f"{g.compute_partition}/{g.memory_partition} "
        f"({len(g.partitions) or 1} device(s))"
    )
    return 0


def cmd_health(args) -> int:
    lib = _lib(args)
    rc = 0
    for g in lib.enumerate():
        h = lib.health_check(g.index)
        print(f"{g.canonical_name}: {h}")
        if h.get("status") != "healthy":
            rc = 1
    if args.probe:
        from . import _hiphealth

        for d in range(_hiphealth.device_count()):
            chk = _hiphealth.mfma_check(d)
            bw = _hiphealth.bandwidth_gbs(d, 256, 5)
            print(f"hip:{d}: mfma_ok={chk['ok']} bandwidth={bw:.0f}GB/s")
            if not chk["ok"]:
                rc = 1
    return rc


def cmd_labels(args) -> int:
    """Preview the node labels the controller would derive from this
    node's devices (gpu.count, architecture, hive, partition modes)."""
    from .controller.manager import labels_for_node

    lib = _lib(args)
    devices = []
    for g in lib.enumerate():
        if g.partitions:
            devices.extend(
                AllocatableDevice.from_partition(g, p).to_device()
                for p in g.partitions
            )
        else:
            devices.append(AllocatableDevice.from_gpu(g).to_device())
    for k, v in sorted(labels_for_node(devices).items()):
        print(f"{k}={v}")
    return 0


def cmd_profile(args) -> int:
    """Fetch an on-demand CPU profile from a running plugin/controller's
    diag server (/debug/profile) — the `go tool pprof` moment for the
    prepare path; output is collapsed-stack (flamegraph.pl-ready)."""
    import urllib.request

    url = (
        f"http://{args.host}:{args.port}/debug/profile"
        f"?seconds={args.seconds}"
    )
    body = urllib.request.urlopen(url, timeout=args.seconds + 30).read()
    sys.stdout.write(body.decode())
    return 0


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("amd-dra-ctl")
    ap.add_argument("--hal", default="amdsmi", choices=["amdsmi", "kfd", "fake"])
    sub = ap.add_subparsers(dest="cmd", required=True)
    sub.add_parser("list")
    sub.add_parser("topology")
    sp = sub.add_parser("slice")
    sp.add_argument(
        "--prospective",
        default="",
        choices=["", "cpx", "dpx", "qpx"],
        help="include prospective partitions + counter sets (what "
        "--prospective-partitions would publish)",
    )
    pp = sub.add_parser("partition")
    pp.add_argument("gpu", type=int)
    pp.add_argument("compute")
    pp.add_argument("memory", nargs="?", default="NPS1")
    sub.add_parser("labels")
    hp = sub.add_parser("health")
    hp.add_argument("--probe", action="store_true", help="also run HIP kernels")
    prof = sub.add_parser("profile")
    prof.add_argument("--host", default="127.0.0.1")
    prof.add_argument("--port", type=int, default=8083)
    prof.add_argument("--seconds", type=float, default=5.0)
    args = ap.parse_args(argv)
    return {
        "list": cmd_list,
        "topology": cmd_topology,
        "slice": cmd_slice,
        "partition": cmd_partition,
        "health": cmd_health,
        "labels": cmd_labels,
        "profile": cmd_profile,
    }[args.cmd](args)


if __name__ == "__main__":
    raise SystemExit(main())
