"""ROCm driver-root discovery — the ``root.go:29-98`` analog.

The reference must locate ``libnvidia-ml.so.1`` / ``nvidia-smi`` under a
(possibly containerized) driver root because NVIDIA userspace must match
the kernel driver. ROCm's contract is different — the amdgpu/KFD kernel
ABI is stable and workload images normally ship their own ROCm — so
injection is OPT-IN (``--rocm-mount``): for slim images with no ROCm
userspace, the plugin can discover the host installation and bind it
read-only into claim containers.

Discovery order (first valid wins):
1. explicit path (``--rocm-mount /path``),
2. ``$ROCM_PATH``,
3. ``<root>/opt/rocm`` (the packaging symlink),
4. highest-versioned ``<root>/opt/rocm-*``.

"Valid" = the directory carries the HIP or HSA runtime under ``lib/``.
"""

from __future__ import annotations

import glob
import logging
import os
import re
from typing import Optional, Tuple

log = logging.getLogger(__name__)

#: at least one of these must exist under <root>/lib for a ROCm userspace
_RUNTIME_GLOBS = ("libamdhip64.so*", "libhsa-runtime64.so*")

_VER_RE = re.compile(r"rocm-(\d+(?:\.\d+)*)")


def _is_rocm_root(path: str) -> bool:
    libdir = os.path.join(path, "lib")
    return any(glob.glob(os.path.join(libdir, g)) for g in _RUNTIME_GLOBS)


def _version_of(path: str) -> str:
    """Best-effort version: .info/version file, else the dir-name suffix."""
    try:
        with open(os.path.join(path, ".info", "version")) as f:
            return f.read().strip()
    except OSError:
        pass
    m = _VER_RE.search(os.path.basename(os.path.realpath(path)))
    return m.group(1) if m else ""


def discover_rocm_root(
    explicit: str = "",
    *,
    host_root: str = "",
    env: Optional[dict] = None,
) -> Optional[Tuple[str, str]]:
    """Locate a ROCm installation; returns (path, version) or None.

    ``host_root`` prefixes the search when the host filesystem is mounted
    at a non-/ path inside the plugin container (the dev-root pattern,
    reference root.go:76-98).
    """
    e = env if env is not None else os.environ
    candidates = []
    if explicit and explicit != "auto":
        candidates.append(explicit)
    else:
        if e.get("ROCM_PATH"):
            candidates.append(e["ROCM_PATH"])
        candidates.append(f"{host_root}/opt/rocm")
        versioned = sorted(
            glob.glob(f"{host_root}/opt/rocm-*"),
            key=lambda p: [
                int(x) for x in (_VER_RE.search(p).group(1).split(".") if _VER_RE.search(p) else ["0"])
            ],
            reverse=True,
        )
        candidates.extend(versioned)
    for c in candidates:
        if c and os.path.isdir(c) and _is_rocm_root(c):
            ver = _version_of(c)
            log.info("discovered ROCm root %s (version %s)", c, ver or "?")
            return c, ver
    if explicit and explicit != "auto":
        log.warning(
            "--rocm-mount %s is not a ROCm installation (no HIP/HSA "
            "runtime under lib/); not injecting",
            explicit,
        )
    return None
