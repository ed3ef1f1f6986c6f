"""Kubernetes client authentication: kubeconfig + in-cluster, client-go parity.

Reference analog: ``pkg/flags/kubeclient.go:30-107``, which inherits the
full authentication surface from client-go's ``clientcmd``. This module
reproduces the parts real clusters actually use:

- inline ``certificate-authority-data`` / ``client-certificate-data`` /
  ``client-key-data`` (what kind and kubeadm admin kubeconfigs ship);
- file-path ``certificate-authority`` / ``client-certificate`` / ``client-key``;
- bearer ``token`` and rotating ``tokenFile``;
- ``username``/``password`` basic auth;
- ``exec`` credential plugins (cloud CLIs: gke-gcloud-auth-plugin, aws eks
  get-token, ...) with ``expirationTimestamp`` caching and support for both
  token and client-certificate outputs;
- in-cluster service-account config with bound-token rotation (the token
  file is re-read when it changes) and a *mandatory* CA bundle — a missing
  ``ca.crt`` is a loud error, never a silent ``verify=False``.

The produced :class:`KubeConnection` is transport-agnostic: it exposes the
server URL, a per-request ``headers()`` provider, an ``ssl_verify()`` object
(path / bool / ``ssl.SSLContext``) and an ``epoch`` that bumps whenever the
TLS material changed so the HTTP client knows to rebuild its transport
(exec plugins may rotate client certificates).
"""

from __future__ import annotations

import base64
import datetime
import json
import logging
import os
import ssl
import subprocess
import tempfile
import threading
from typing import Dict, List, Optional, Tuple

import yaml

log = logging.getLogger(__name__)

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"
SA_TOKEN = f"{SA_DIR}/token"
SA_CA = f"{SA_DIR}/ca.crt"

#: minimum remaining lifetime before an exec credential is re-fetched
_EXEC_EXPIRY_SLACK_S = 60.0
#: token files are re-stat'd at most this often
_TOKEN_RECHECK_S = 60.0


class KubeAuthError(RuntimeError):
    """Config is unusable: unreadable kubeconfig, unsupported auth, bad TLS."""


# ---------------------------------------------------------------------------
# credential providers
# ---------------------------------------------------------------------------
class _Static:
    def __init__(self, headers: Dict[str, str]):
        self._headers = dict(headers)

    def headers(self) -> Dict[str, str]:
        return dict(self._headers)


class _TokenFile:
    """Bearer token re-read when the file changes (bound-SA rotation)."""

    def __init__(self, path: str, recheck_s: float = _TOKEN_RECHECK_S):
        self.path = path
        self.recheck_s = recheck_s
        self._lock = threading.Lock()
        self._token = ""
        self._mtime = 0.0
        self._checked = 0.0
        self._read(force=True)

    def _read(self, force: bool = False) -> None:
        import time

        now = time.monotonic()
        if not force and now - self._checked < self.recheck_s:
            return
        self._checked = now
        try:
            mtime = os.stat(self.path).st_mtime
        except OSError as e:
            if self._token:
                return  # keep the cached token; file may be mid-rotation
            raise KubeAuthError(f"token file unreadable: {self.path}: {e}")
        if force or mtime != self._mtime:
            with open(self.path) as f:
                self._token = f.read().strip()
            self._mtime = mtime

    def headers(self) -> Dict[str, str]:
        with self._lock:
            self._read()
            return {"Authorization": f"Bearer {self._token}"}


class _Exec:
    """ExecCredential plugin (client.authentication.k8s.io/v1 + v1beta1).

    Runs the configured command, parses the ExecCredential status, and
    caches it until ``expirationTimestamp`` (minus slack). Supports token
    output and client-certificate output; for the latter the PEM data is
    written to private temp files and the connection epoch is bumped so the
    transport reloads its TLS context.
    """

    def __init__(self, spec: dict, conn: "KubeConnection"):
        self.command = spec.get("command")
        if not self.command or not isinstance(self.command, str):
            raise KubeAuthError("exec credential plugin without command")
        self.args: List[str] = list(spec.get("args") or [])
        env_list = spec.get("env") or []
        if not isinstance(env_list, list):
            raise KubeAuthError("exec.env must be a list")
        self.env_add = {}
        for e in env_list:
            if isinstance(e, dict) and isinstance(e.get("name"), str):
                self.env_add[e["name"]] = str(e.get("value", ""))
        self.api_version = spec.get(
            "apiVersion", "client.authentication.k8s.io/v1beta1"
        )
        self.provide_cluster_info = bool(spec.get("provideClusterInfo"))
        self.conn = conn
        self._lock = threading.Lock()
        self._token: Optional[str] = None
        self._expiry: Optional[datetime.datetime] = None

    def _expired(self) -> bool:
        if self._token is None and self.conn.client_cert is None:
            return True
        if self._expiry is None:
            return self._token is None
        now = datetime.datetime.now(datetime.timezone.utc)
        return (self._expiry - now).total_seconds() < _EXEC_EXPIRY_SLACK_S

    def _refresh(self) -> None:
        env = dict(os.environ)
        env.update(self.env_add)
        exec_info: dict = {
            "apiVersion": self.api_version,
            "kind": "ExecCredential",
            "spec": {"interactive": False},
        }
        if self.provide_cluster_info:
            exec_info["spec"]["cluster"] = {
                "server": self.conn.server,
            }
        env["KUBERNETES_EXEC_INFO"] = json.dumps(exec_info)
        try:
            out = subprocess.run(
                [self.command, *self.args],
                env=env,
                capture_output=True,
                timeout=60,
                check=True,
            ).stdout
        except FileNotFoundError:
            raise KubeAuthError(
                f"exec credential plugin not found: {self.command}"
            )
        except subprocess.CalledProcessError as e:
            raise KubeAuthError(
                f"exec credential plugin failed ({self.command}): "
                f"{e.stderr.decode(errors='replace').strip()[:500]}"
            )
        except subprocess.TimeoutExpired:
            raise KubeAuthError(
                f"exec credential plugin timed out: {self.command}"
            )
        try:
            cred = json.loads(out)
            status = cred["status"]
        except (json.JSONDecodeError, KeyError) as e:
            raise KubeAuthError(
                f"exec credential plugin emitted bad ExecCredential: {e}"
            )
        self._token = status.get("token")
        exp = status.get("expirationTimestamp")
        self._expiry = _parse_rfc3339(exp) if exp else None
        cert_data = status.get("clientCertificateData")
        key_data = status.get("clientKeyData")
        if cert_data and key_data:
            self.conn._set_client_cert_data(cert_data, key_data)

    def headers(self) -> Dict[str, str]:
        with self._lock:
            if self._expired():
                self._refresh()
            if self._token:
                return {"Authorization": f"Bearer {self._token}"}
            return {}


def _parse_rfc3339(s: str) -> datetime.datetime:
    s = s.replace("Z", "+00:00")
    dt = datetime.datetime.fromisoformat(s)
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=datetime.timezone.utc)
    return dt


# ---------------------------------------------------------------------------
# connection
# ---------------------------------------------------------------------------
class KubeConnection:
    """Resolved cluster endpoint + authentication material."""

    def __init__(
        self,
        server: str,
        *,
        insecure_skip_verify: bool = False,
        ca_file: Optional[str] = None,
        ca_data: Optional[bytes] = None,
    ):
        if not server or not isinstance(server, str):
            raise KubeAuthError("cluster has no server URL")
        self.server = server.rstrip("/")
        self.insecure = insecure_skip_verify
        self.ca_file = ca_file
        self.ca_data = ca_data
        #: (cert_path, key_path) for client TLS, if any
        self.client_cert: Optional[Tuple[str, str]] = None
        #: bumps whenever TLS material changes -> transport rebuild
        self.epoch = 0
        self._provider: Optional[object] = None
        self._tmpdir: Optional[str] = None

    # -- TLS -------------------------------------------------------------
    def _tmp(self) -> str:
        if self._tmpdir is None:
            self._tmpdir = tempfile.mkdtemp(prefix="amd-dra-kube-")
            os.chmod(self._tmpdir, 0o700)
        return self._tmpdir

    def _write_secret(self, name: str, data: bytes) -> str:
        path = os.path.join(self._tmp(), name)
        fd = os.open(path, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, 0o600)
        with os.fdopen(fd, "wb") as f:
            f.write(data)
        return path

    def _set_client_cert_data(self, cert_pem: str, key_pem: str) -> None:
        cert = self._write_secret("client.crt", cert_pem.encode())
        key = self._write_secret("client.key", key_pem.encode())
        if self.client_cert != (cert, key):
            self.client_cert = (cert, key)
        self.epoch += 1  # PEM contents may differ even at same paths

    def set_client_cert_files(self, cert_path: str, key_path: str) -> None:
        for p in (cert_path, key_path):
            if not os.path.exists(p):
                raise KubeAuthError(f"client certificate file missing: {p}")
        self.client_cert = (cert_path, key_path)
        self.epoch += 1

    def ssl_verify(self):
        """Build the httpx ``verify`` argument for the current material."""
        if not self.server.startswith("https"):
            return False
        if self.insecure:
            if self.client_cert:
                ctx = ssl._create_unverified_context()  # noqa: SLF001
                ctx.load_cert_chain(*self.client_cert)
                return ctx
            return False
        if self.ca_file or self.ca_data or self.client_cert:
            kwargs: dict = {}
            if self.ca_file:
                if not os.path.exists(self.ca_file):
                    raise KubeAuthError(
                        f"certificate-authority file missing: {self.ca_file}"
                    )
                kwargs["cafile"] = self.ca_file
            elif self.ca_data:
                kwargs["cadata"] = self.ca_data.decode()
            try:
                ctx = ssl.create_default_context(**kwargs)
            except ssl.SSLError as e:
                raise KubeAuthError(f"bad certificate-authority data: {e}")
            if self.client_cert:
                try:
                    ctx.load_cert_chain(*self.client_cert)
                except ssl.SSLError as e:
                    raise KubeAuthError(f"bad client certificate/key: {e}")
            return ctx
        return True  # system trust roots

    # -- auth ------------------------------------------------------------
    def set_provider(self, provider: object) -> None:
        self._provider = provider

    def headers(self) -> Dict[str, str]:
        if self._provider is None:
            return {}
        return self._provider.headers()  # type: ignore[attr-defined]


# ---------------------------------------------------------------------------
# loaders
# ---------------------------------------------------------------------------
def load_in_cluster(env: Optional[dict] = None) -> KubeConnection:
    """In-cluster config: SA token + mandatory CA (kubeclient.go:97-101).

    A missing ``ca.crt`` raises instead of silently disabling TLS
    verification (round-1 advisor finding)."""
    e = env if env is not None else os.environ
    host = e.get("KUBERNETES_SERVICE_HOST")
    port = e.get("KUBERNETES_SERVICE_PORT", "443")
    if not host:
        raise KubeAuthError(
            "not running in-cluster (KUBERNETES_SERVICE_HOST unset) and no "
            "kubeconfig given"
        )
    ca = e.get("AMD_DRA_SA_CA", SA_CA)
    token = e.get("AMD_DRA_SA_TOKEN", SA_TOKEN)
    if not os.path.exists(ca):
        raise KubeAuthError(
            f"in-cluster CA bundle missing at {ca}; refusing to talk to the "
            "apiserver without TLS verification (mount the serviceaccount "
            "volume, or pass --kubeconfig)"
        )
    if ":" in host and not host.startswith("["):
        host = f"[{host}]"  # IPv6
    conn = KubeConnection(f"https://{host}:{port}", ca_file=ca)
    conn.set_provider(_TokenFile(token))
    return conn


def _load_one(path: str) -> dict:
    try:
        with open(path) as f:
            cfg = yaml.safe_load(f) or {}
    except OSError as e:
        raise KubeAuthError(f"kubeconfig unreadable: {path}: {e}")
    except yaml.YAMLError as e:
        raise KubeAuthError(f"kubeconfig is not valid YAML: {path}: {e}")
    if not isinstance(cfg, dict):
        raise KubeAuthError(
            f"kubeconfig root must be a mapping, got {type(cfg).__name__}"
        )
    return cfg


def _merge_kubeconfigs(paths: List[str]) -> tuple:
    """clientcmd KUBECONFIG-list merge: for each named entry the FIRST
    file wins; current-context comes from the first file that sets one.
    Returns (merged_cfg, base_dir_by_entry_name) so relative file paths
    still resolve against the file that defined them."""
    merged: dict = {"contexts": [], "clusters": [], "users": []}
    bases: Dict[tuple, str] = {}
    for p in paths:
        cfg = _load_one(p)
        base = os.path.dirname(os.path.abspath(p))
        if not merged.get("current-context") and cfg.get("current-context"):
            merged["current-context"] = cfg["current-context"]
        for kind in ("contexts", "clusters", "users"):
            have = {
                e.get("name")
                for e in merged[kind]
                if isinstance(e, dict)
            }
            entries = cfg.get(kind) or []
            if not isinstance(entries, list):
                raise KubeAuthError(
                    f"kubeconfig {p}: {kind} must be a list"
                )
            for e in entries:
                if isinstance(e, dict) and e.get("name") not in have:
                    merged[kind].append(e)
                    bases[(kind, e.get("name"))] = base
    return merged, bases


def load_kubeconfig(
    path: str, context: Optional[str] = None
) -> KubeConnection:
    """Parse a kubeconfig into a connection (clientcmd analog).

    Supports inline *-data fields, file paths (resolved relative to the
    kubeconfig), token/tokenFile, basic auth, exec credential plugins,
    and a colon-separated KUBECONFIG path list (first-wins merge).
    """
    if ":" in path and not os.path.exists(path):
        paths = [p for p in path.split(":") if p]
        cfg, bases = _merge_kubeconfigs(paths)
        return _connection_from_cfg(cfg, context, bases)
    cfg = _load_one(path)

    base = os.path.dirname(os.path.abspath(path))
    return _connection_from_cfg(cfg, context, None, default_base=base)


def _connection_from_cfg(
    cfg: dict,
    context: Optional[str],
    bases: Optional[Dict[tuple, str]],
    default_base: str = "",
) -> KubeConnection:
    ctx_name = context or cfg.get("current-context")
    if not ctx_name or not isinstance(ctx_name, str):
        raise KubeAuthError("kubeconfig has no current-context")
    ctx = _named(cfg.get("contexts"), ctx_name, "context")
    cluster = _named(cfg.get("clusters"), ctx.get("cluster"), "cluster")
    user = _named(cfg.get("users"), ctx.get("user"), "user") if ctx.get("user") else {}

    cluster_base = (
        bases.get(("clusters", ctx.get("cluster")), default_base)
        if bases
        else default_base
    )
    user_base = (
        bases.get(("users", ctx.get("user")), default_base)
        if bases
        else default_base
    )

    def _respath(p, base) -> Optional[str]:
        if not p:
            return None
        if not isinstance(p, str):
            raise KubeAuthError("kubeconfig file path fields must be strings")
        return p if os.path.isabs(p) else os.path.join(base, p)

    def respath(p) -> Optional[str]:
        return _respath(p, user_base)

    def respath_cluster(p) -> Optional[str]:
        return _respath(p, cluster_base)

    ca_data = None
    if cluster.get("certificate-authority-data"):
        ca_data = _b64(cluster["certificate-authority-data"], "certificate-authority-data")
    conn = KubeConnection(
        cluster.get("server", ""),
        insecure_skip_verify=bool(cluster.get("insecure-skip-tls-verify")),
        ca_file=respath_cluster(cluster.get("certificate-authority")),
        ca_data=ca_data,
    )

    # client TLS
    cert_data = user.get("client-certificate-data")
    key_data = user.get("client-key-data")
    if cert_data and key_data:
        conn._set_client_cert_data(
            _b64(cert_data, "client-certificate-data").decode(),
            _b64(key_data, "client-key-data").decode(),
        )
    elif user.get("client-certificate") and user.get("client-key"):
        conn.set_client_cert_files(
            respath(user["client-certificate"]),
            respath(user["client-key"]),
        )

    # bearer / basic / exec
    if user.get("token"):
        if not isinstance(user["token"], str):
            raise KubeAuthError("user.token must be a string")
        conn.set_provider(
            _Static({"Authorization": f"Bearer {user['token']}"})
        )
    elif user.get("tokenFile"):
        conn.set_provider(_TokenFile(respath(user["tokenFile"])))
    elif user.get("username") is not None and user.get("password") is not None:
        basic = base64.b64encode(
            f"{user['username']}:{user['password']}".encode()
        ).decode()
        conn.set_provider(_Static({"Authorization": f"Basic {basic}"}))
    elif user.get("exec"):
        if not isinstance(user["exec"], dict):
            raise KubeAuthError("user.exec must be a mapping")
        conn.set_provider(_Exec(user["exec"], conn))
    elif user.get("auth-provider"):
        raise KubeAuthError(
            "legacy auth-provider kubeconfigs are not supported (removed "
            "from client-go too); migrate to an exec credential plugin"
        )
    # else: client-cert-only or anonymous — both valid

    return conn


def _named(entries, name: Optional[str], kind: str) -> dict:
    if entries is not None and not isinstance(entries, list):
        raise KubeAuthError(f"kubeconfig {kind}s must be a list")
    for e in entries or []:
        if isinstance(e, dict) and e.get("name") == name:
            body = e.get(kind) or {}
            if not isinstance(body, dict):
                raise KubeAuthError(
                    f"kubeconfig {kind} {name!r} body must be a mapping"
                )
            return body
    raise KubeAuthError(f"kubeconfig {kind} {name!r} not found")


def _b64(data, what: str) -> bytes:
    if not isinstance(data, str):
        raise KubeAuthError(f"{what} must be a base64 string")
    try:
        return base64.b64decode(data)
    except Exception as e:
        raise KubeAuthError(f"bad base64 in {what}: {e}")
