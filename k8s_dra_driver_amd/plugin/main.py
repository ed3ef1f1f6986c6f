"""Kubelet plugin CLI entry point (reference cmd/nvidia-dra-plugin/main.go).

Every flag mirrors an env var (reference main.go:73-123); Helm wires them
from chart values. Runs until SIGTERM/SIGINT, then unpublishes slices and
removes sockets.
"""

from __future__ import annotations

import argparse
import logging
import os
import signal
import threading

from .. import DRIVER_NAME
from ..hal import new_device_lib
from ..kube.client import InMemoryKube
from ..metrics.prom import PluginMetrics
from .driver import Driver
from .server import PluginServer

log = logging.getLogger(__name__)

DEFAULT_PLUGIN_ROOT = "/var/lib/kubelet/plugins"
DEFAULT_REGISTRY = "/var/lib/kubelet/plugins_registry"
DEFAULT_CDI_ROOT = "/var/run/cdi"


def _env(name: str, default: str = "") -> str:
    return os.environ.get(name, default)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser("amd-dra-kubeletplugin")
    from ..version import version_string

    p.add_argument(
        "--version",
        action="version",
        version=f"%(prog)s {version_string()}",
    )
    p.add_argument(
        "--node-name",
        default=_env("NODE_NAME", os.uname().nodename),
        help="name of the node this plugin runs on",
    )
    p.add_argument(
        "--hal",
        default=_env("DRA_HAL", "amdsmi"),
        choices=["amdsmi", "kfd", "fake"],
        help="hardware backend (fake = 8xMI355X model for dev clusters)",
    )
    p.add_argument(
        "--cdi-root", default=_env("CDI_ROOT", DEFAULT_CDI_ROOT)
    )
    p.add_argument(
        "--device-classes",
        default=_env("DEVICE_CLASSES", "gpu,partition"),
        help="comma list of device kinds to publish (gpu,partition) — "
        "subsystem gating parity with the reference",
    )
    p.add_argument(
        "--plugin-registration-path",
        default=_env("PLUGIN_REGISTRATION_PATH", DEFAULT_REGISTRY),
    )
    p.add_argument(
        "--plugin-path",
        default=_env(
            "PLUGIN_PATH", os.path.join(DEFAULT_PLUGIN_ROOT, DRIVER_NAME)
        ),
    )
    p.add_argument(
        "--kubeconfig",
        default=_env("KUBECONFIG", ""),
        help="kubeconfig path ('' = in-cluster; 'memory' = in-memory fake)",
    )
    p.add_argument(
        "--kube-api-qps", type=float, default=float(_env("KUBE_API_QPS", "50"))
    )
    p.add_argument(
        "--kube-api-burst", type=int, default=int(_env("KUBE_API_BURST", "100"))
    )
    p.add_argument(
        "--diag-port",
        type=int,
        default=int(_env("DIAG_PORT", "0")),
        help="serve /healthz + /debug diagnostics on this port (0 = off)",
    )
    p.add_argument(
        "--metrics-port",
        type=int,
        default=int(_env("METRICS_PORT", "0")),
        help="serve Prometheus /metrics on this port (0 = disabled)",
    )
    p.add_argument(
        "--prospective-partitions",
        default=_env("PROSPECTIVE_PARTITIONS", ""),
        choices=["", "cpx", "dpx", "qpx"],
        help="publish prospective partitions of this mode with DRA "
        "sharedCounters (K8s 1.33 partitionable devices) so the default "
        "scheduler can allocate them; prepare carves on demand — "
        "scheduler-driven dynamic partitioning (needs resource.k8s.io "
        "v1beta2+)",
    )
    p.add_argument(
        "--rocm-mount",
        default=_env("ROCM_MOUNT", ""),
        help="inject host ROCm userspace into claim containers: a path, "
        "'auto' (discover ROCM_PATH / /opt/rocm / /opt/rocm-*), or empty "
        "= off (images bring their own ROCm; root.go:29-98 analog)",
    )
    p.add_argument(
        "--dev-root",
        default=_env("DEV_ROOT", ""),
        help="prefix where the host filesystem is mounted inside this "
        "container (affects /dev paths in CDI specs and ROCm discovery)",
    )
    p.add_argument(
        "--gpu-indices",
        default=_env("GPU_INDICES", ""),
        help="comma-separated GPU indices this plugin instance manages "
        "(empty = all) — the nvkind multi-node-simulation analog: split "
        "one box's GPUs across several 'nodes'",
    )
    p.add_argument(
        "--shared-enforcement",
        default=_env("SHARED_ENFORCEMENT", "warn"),
        choices=["off", "warn", "kill"],
        help="shared-GPU isolation enforcement for containers that strip "
        "or alter their CU mask: warn = Warning event + metric, kill = "
        "SIGKILL the offending process (sharing.go:211-221 parity)",
    )
    p.add_argument("-v", "--verbosity", type=int, default=int(_env("LOG_LEVEL", "1")))
    p.add_argument(
        "--logging-format",
        default=_env("LOGGING_FORMAT", "text"),
        choices=["text", "json"],
        help="log output format (logsapi parity)",
    )
    return p


def make_kube_client(args):
    if args.kubeconfig == "memory":
        return InMemoryKube()
    from ..kube.http_kube import HttpKube

    return HttpKube(
        kubeconfig=args.kubeconfig or None,
        qps=args.kube_api_qps,
        burst=args.kube_api_burst,
    )


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    from ..utils.log import setup_logging

    setup_logging(args.verbosity, json_format=args.logging_format == "json")
    lib = new_device_lib(args.hal)
    lib.open()
    kube = make_kube_client(args)
    metrics = PluginMetrics()
    if args.metrics_port:
        metrics.serve(args.metrics_port)
    diag = None
    if args.diag_port:
        from ..utils.diag import DiagServer

        diag = DiagServer(args.diag_port)
        diag.start()
    driver = Driver(
        lib,
        kube,
        node_name=args.node_name,
        cdi_root=args.cdi_root,
        checkpoint_root=os.path.join(args.plugin_path, "state"),
        metrics=metrics,
        device_kinds=[s.strip() for s in args.device_classes.split(",") if s.strip()],
        shared_enforcement=args.shared_enforcement,
        rocm_mount=args.rocm_mount,
        dev_root=args.dev_root,
        prospective_partitions=args.prospective_partitions,
        gpu_indices=(
            [int(s) for s in args.gpu_indices.split(",") if s.strip()]
            if args.gpu_indices
            else None
        ),
    )
    driver.startup()
    driver.health.start()  # failure detection -> slice self-healing
    if driver.enforcer is not None:
        driver.enforcer.start()  # shared-GPU isolation watchdog
    server = PluginServer(
        driver,
        plugin_dir=args.plugin_path,
        registry_dir=args.plugin_registration_path,
    )
    server.start()
    server.start_socket_watchdog()

    # periodic orphan cleanup (reference TODO parity, driver.go:156-168)
    stop = threading.Event()

    def cleanup_loop():
        while not stop.wait(300):
            try:
                driver.cleanup_orphans()
            except Exception:
                log.exception("orphan cleanup pass failed")

    threading.Thread(target=cleanup_loop, name="orphan-gc", daemon=True).start()
    for sig in (signal.SIGTERM, signal.SIGINT):
        signal.signal(sig, lambda *_: stop.set())
    log.info(
        "amd-dra-kubeletplugin ready: node=%s hal=%s", args.node_name, args.hal
    )
    stop.wait()
    if diag is not None:
        diag.stop()
    driver.health.stop()
    server.stop()
    driver.shutdown()
    lib.close()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
