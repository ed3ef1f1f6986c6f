"""Diagnostics endpoint tests."""

import json
import urllib.request

import pytest

from k8s_dra_driver_amd.utils.diag import DiagServer


@pytest.fixture
def diag_url():
    srv = DiagServer(0, host="127.0.0.1")
    srv.start()
    yield f"http://127.0.0.1:{srv.port}"
    srv.stop()


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


def test_profile_endpoint_collapsed_stacks(diag_url):
    """pprof `profile` analog: on-demand sampling CPU
    profile in collapsed-stack format, covering worker threads."""
    import threading
    import urllib.request

    stop = threading.Event()

    def busy_loop_marker():
        while not stop.is_set():
            sum(i * i for i in range(2000))

    t = threading.Thread(target=busy_loop_marker, name="busy", daemon=True)
    t.start()
    try:
        body = (
            urllib.request.urlopen(f"{diag_url}/debug/profile?seconds=0.4")
            .read()
            .decode()
        )
    finally:
        stop.set()
        t.join()
    assert body.startswith("# cpu profile")
    assert "busy_loop_marker" in body  # the worker thread was sampled


def test_profile_endpoint_json(diag_url):
    import json as _json
    import urllib.request

    body = urllib.request.urlopen(
        f"{diag_url}/debug/profile?seconds=0.2&format=json"
    ).read()
    d = _json.loads(body)
    assert d["samples"] > 0
    assert isinstance(d["stacks"], dict)
