"""Diagnostics HTTP endpoint — the pprof-analog (reference
cmd/nvidia-dra-controller/main.go:216-224 mounts Go pprof; the Python
equivalent exposes liveness + thread stack dumps + gc stats)."""

from __future__ import annotations

import gc
import json
import sys
import threading
import traceback
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional


class _Handler(BaseHTTPRequestHandler):
    def log_message(self, *a):  # quiet
        pass

    def _send(self, code: int, body: str, ctype="text/plain"):
        data = body.encode()
        self.send_response(code)
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_GET(self):
        if self.path == "/healthz":
            self._send(200, "ok")
        elif self.path == "/debug/threads":
            frames = sys._current_frames()
            out = []
            for t in threading.enumerate():
                frame = frames.get(t.ident)
                stack = (
                    "".join(traceback.format_stack(frame)) if frame else "<no frame>"
                )
                out.append(f"--- {t.name} (daemon={t.daemon})\n{stack}")
            self._send(200, "\n".join(out))
        elif self.path == "/debug/gc":
            self._send(
                200,
                json.dumps(
                    {
                        "counts": gc.get_count(),
                        "threshold": gc.get_threshold(),
                        "objects": len(gc.get_objects()),
                        "threads": len(threading.enumerate()),
                    }
                ),
                "application/json",
            )
        else:
            self._send(404, "not found; try /healthz /debug/threads /debug/gc")


class DiagServer:
    def __init__(self, port: int, host: str = "0.0.0.0"):
        self._server = ThreadingHTTPServer((host, port), _Handler)
        self._thread: Optional[threading.Thread] = None

    @property
    def port(self) -> int:
        return self._server.server_address[1]

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="diag-http", daemon=True
        )
        self._thread.start()

    def stop(self) -> None:
        self._server.shutdown()
        if self._thread:
            self._thread.join(timeout=5)
