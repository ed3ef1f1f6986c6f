"""Cluster controller CLI (reference cmd/nvidia-dra-controller/main.go)."""

from __future__ import annotations

import argparse
import logging
import os
import signal
import threading

from ..metrics.prom import PluginMetrics

log = logging.getLogger(__name__)


def _env(name: str, default: str = "") -> str:
    return os.environ.get(name, default)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser("amd-dra-controller")
    from ..version import version_string

    p.add_argument(
        "--version",
        action="version",
        version=f"%(prog)s {version_string()}",
    )
    p.add_argument("--kubeconfig", default=_env("KUBECONFIG", ""))
    p.add_argument(
        "--poll-interval", type=float, default=float(_env("POLL_INTERVAL", "10"))
    )
    p.add_argument(
        "--metrics-port", type=int, default=int(_env("METRICS_PORT", "8085"))
    )
    p.add_argument(
        "--diag-port",
        type=int,
        default=int(_env("DIAG_PORT", "0")),
        help="serve /healthz + /debug diagnostics on this port (0 = off)",
    )
    p.add_argument(
        "--allocate-claims",
        default=_env("ALLOCATE_CLAIMS", "false").lower() == "true",
        action="store_true",
        help="allocate pending ResourceClaims with the topology-aware "
        "allocator (leave off when kube-scheduler performs DRA allocation)",
    )
    p.add_argument(
        "--manage-node-labels",
        default=_env("MANAGE_NODE_LABELS", "true").lower() == "true",
        action="store_true",
    )
    p.add_argument("-v", "--verbosity", type=int, default=int(_env("LOG_LEVEL", "1")))
    p.add_argument(
        "--logging-format",
        default=_env("LOGGING_FORMAT", "text"),
        choices=["text", "json"],
        help="log output format (logsapi parity)",
    )
    return p


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    from ..utils.log import setup_logging

    setup_logging(args.verbosity, json_format=args.logging_format == "json")
    if args.kubeconfig == "memory":
        from ..kube.client import InMemoryKube

        kube = InMemoryKube()
    else:
        from ..kube.http_kube import HttpKube

        kube = HttpKube(kubeconfig=args.kubeconfig or None)

    from .manager import ControllerManager

    metrics = PluginMetrics()
    if args.metrics_port:
        metrics.serve(args.metrics_port)
    diag = None
    if args.diag_port:
        from ..utils.diag import DiagServer

        diag = DiagServer(args.diag_port)
        diag.start()

    mgr = ControllerManager(
        kube,
        poll_interval=args.poll_interval,
        manage_labels=args.manage_node_labels,
        allocate_claims=args.allocate_claims,
    )
    mgr.start()
    stop = threading.Event()
    for sig in (signal.SIGTERM, signal.SIGINT):
        signal.signal(sig, lambda *_: stop.set())
    log.info("amd-dra-controller ready")
    stop.wait()
    if diag is not None:
        diag.stop()
    mgr.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
