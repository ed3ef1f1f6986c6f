"""Diagnostics endpoint tests."""

import json
import urllib.request

from k8s_dra_driver_amd.utils.diag import DiagServer


def test_diag_endpoints():
    srv = DiagServer(0, host="127.0.0.1")
    srv.start()
    base = f"http://127.0.0.1:{srv.port}"
    try:
        assert urllib.request.urlopen(f"{base}/healthz").read() == b"ok"
        threads = urllib.request.urlopen(f"{base}/debug/threads").read().decode()
        assert "diag-http" in threads or "MainThread" in threads
        gc_info = json.loads(
            urllib.request.urlopen(f"{base}/debug/gc").read()
        )
        assert gc_info["threads"] >= 1
    finally:
        srv.stop()
