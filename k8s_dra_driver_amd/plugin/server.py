"""gRPC servers: DRA plugin service + kubelet plugin registration.

The socket/registration lifecycle the reference inherits from the vendored
``kubeletplugin`` helper (``draplugin.go:280-350``): a DRA service socket at
``<plugin-dir>/plugin.sock`` and a registration socket at
``<registry-dir>/<driver-name>.sock`` where kubelet's pluginwatcher calls
``Registration/GetInfo``. Three DRA API versions are served for kubelet
version negotiation — dra.v1 (GA, K8s 1.34), v1beta1 DRAPlugin and
v1alpha4 Node — one generation past the reference's pair
(``draplugin.go:342-350``).
"""

from __future__ import annotations

import logging
import os
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import List, Optional

import grpc

from .. import DRIVER_NAME
from .driver import ClaimRef, Driver
from .proto import (
    DRA_VERSION_V1,
    DRA_VERSION_V1ALPHA4,
    DRA_VERSION_V1BETA1,
    REGISTRATION,
    V1,
    V1ALPHA4,
    V1BETA1,
    DraMessages,
)

log = logging.getLogger(__name__)


def _dra_handlers(driver: Driver, msgs: DraMessages) -> grpc.GenericRpcHandler:
    def node_prepare(request, context):
        refs = [
            ClaimRef(namespace=c.namespace, name=c.name, uid=c.uid)
            for c in request.claims
        ]
        results = driver.node_prepare_resources(refs)
        resp = msgs.NodePrepareResourcesResponse()
        for uid, res in results.items():
            one = msgs.NodePrepareResourceResponse()
            if res.error:
                one.error = res.error
            else:
                for d in res.devices:
                    dev = one.devices.add()
                    dev.request_names.extend(d["request_names"])
                    dev.pool_name = d["pool_name"]
                    dev.device_name = d["device_name"]
                    dev.cdi_device_ids.extend(d["cdi_device_ids"])
            resp.claims[uid].CopyFrom(one)
        return resp

    def node_unprepare(request, context):
        refs = [
            ClaimRef(namespace=c.namespace, name=c.name, uid=c.uid)
            for c in request.claims
        ]
        results = driver.node_unprepare_resources(refs)
        resp = msgs.NodeUnprepareResourcesResponse()
        for uid, res in results.items():
            one = msgs.NodeUnprepareResourceResponse()
            if res.error:
                one.error = res.error
            resp.claims[uid].CopyFrom(one)
        return resp

    handlers = {
        "NodePrepareResources": grpc.unary_unary_rpc_method_handler(
            node_prepare,
            request_deserializer=msgs.NodePrepareResourcesRequest.FromString,
            response_serializer=lambda m: m.SerializeToString(),
        ),
        "NodeUnprepareResources": grpc.unary_unary_rpc_method_handler(
            node_unprepare,
            request_deserializer=msgs.NodeUnprepareResourcesRequest.FromString,
            response_serializer=lambda m: m.SerializeToString(),
        ),
    }
    return grpc.method_handlers_generic_handler(msgs.service_name, handlers)


def _registration_handlers(
    plugin_endpoint: str, supported_versions: List[str]
) -> grpc.GenericRpcHandler:
    def get_info(request, context):
        return REGISTRATION.PluginInfo(
            type="DRAPlugin",
            name=DRIVER_NAME,
            endpoint=plugin_endpoint,
            supported_versions=supported_versions,
        )

    def notify(request, context):
        if request.plugin_registered:
            log.info("kubelet registered plugin %s", DRIVER_NAME)
        else:
            log.error("kubelet registration failed: %s", request.error)
        return REGISTRATION.RegistrationStatusResponse()

    handlers = {
        "GetInfo": grpc.unary_unary_rpc_method_handler(
            get_info,
            request_deserializer=REGISTRATION.InfoRequest.FromString,
            response_serializer=lambda m: m.SerializeToString(),
        ),
        "NotifyRegistrationStatus": grpc.unary_unary_rpc_method_handler(
            notify,
            request_deserializer=REGISTRATION.RegistrationStatus.FromString,
            response_serializer=lambda m: m.SerializeToString(),
        ),
    }
    return grpc.method_handlers_generic_handler(REGISTRATION.service_name, handlers)


class PluginServer:
    """Owns the gRPC server bound to the plugin + registration sockets."""

    def __init__(
        self,
        driver: Driver,
        *,
        plugin_dir: str,
        registry_dir: Optional[str] = None,
        grpc_workers: int = 16,
    ):
        self.driver = driver
        self.plugin_dir = plugin_dir
        self.registry_dir = registry_dir
        self.plugin_sock = os.path.join(plugin_dir, "plugin.sock")
        self.registry_sock = (
            os.path.join(registry_dir, f"{DRIVER_NAME}.sock")
            if registry_dir
            else None
        )
        self._server: Optional[grpc.Server] = None
        self._grpc_workers = grpc_workers
        self._watchdog_stop = threading.Event()
        self._watchdog: Optional[threading.Thread] = None

    def start(self) -> None:
        os.makedirs(self.plugin_dir, exist_ok=True)
        for sock in filter(None, [self.plugin_sock, self.registry_sock]):
            try:
                os.unlink(sock)
            except FileNotFoundError:
                pass
        server = grpc.server(
            ThreadPoolExecutor(
                max_workers=self._grpc_workers, thread_name_prefix="grpc"
            ),
            options=[
                # prepare latency is the north-star metric; kubelet is a
                # single local client over a unix socket
                ("grpc.optimization_target", "latency"),
            ],
        )
        server.add_generic_rpc_handlers(
            (
                _dra_handlers(self.driver, V1),
                _dra_handlers(self.driver, V1BETA1),
                _dra_handlers(self.driver, V1ALPHA4),
                _registration_handlers(
                    self.plugin_sock,
                    [
                        DRA_VERSION_V1,
                        DRA_VERSION_V1BETA1,
                        DRA_VERSION_V1ALPHA4,
                    ],
                ),
            )
        )
        server.add_insecure_port(f"unix://{self.plugin_sock}")
        if self.registry_sock:
            os.makedirs(self.registry_dir, exist_ok=True)
            server.add_insecure_port(f"unix://{self.registry_sock}")
        server.start()
        self._server = server
        log.info("DRA plugin serving on %s", self.plugin_sock)

    def start_socket_watchdog(self, interval_s: float = 10.0) -> None:
        """Re-bind when kubelet wipes the plugin/registry sockets (e.g. a
        kubelet restart recreating plugins_registry). The vendored
        kubeletplugin gets this from the registration protocol retry; here
        a stat loop restarts the gRPC server on socket loss — prepared
        claims are unaffected (state is on disk + in DeviceState)."""
        self._watchdog_stop.clear()  # allow restart after a stop()

        def run():
            while not self._watchdog_stop.wait(interval_s):
                missing = [
                    s
                    for s in filter(
                        None, [self.plugin_sock, self.registry_sock]
                    )
                    if not os.path.exists(s)
                ]
                if missing and self._server is not None:
                    log.warning(
                        "socket(s) %s vanished (kubelet restart?); "
                        "re-binding",
                        missing,
                    )
                    try:
                        self._server.stop(grace=1.0).wait()
                        self._server = None
                        self.start()
                    except Exception:
                        log.exception("socket re-bind failed; will retry")

        self._watchdog = threading.Thread(
            target=run, name="socket-watchdog", daemon=True
        )
        self._watchdog.start()

    def stop(self, grace: float = 2.0) -> None:
        self._watchdog_stop.set()
        if self._watchdog is not None:
            self._watchdog.join(timeout=5)
        if self._server is not None:
            self._server.stop(grace).wait()
            self._server = None
        for sock in filter(None, [self.plugin_sock, self.registry_sock]):
            try:
                os.unlink(sock)
            except FileNotFoundError:
                pass
