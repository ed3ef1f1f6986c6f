"""Version stamping (reference parity: internal/info/version.go:22-43).

The reference stamps version/commit at build time via ``-ldflags -X``. Here the
same information is resolved at import time: the packaged version constant,
plus a best-effort git describe when running from a checkout.
"""

from __future__ import annotations

import os
import subprocess

__version__ = "0.1.0"


def git_commit() -> str:
    """Best-effort git commit of the running checkout ('' outside a repo)."""
    try:
        out = subprocess.run(
            ["git", "rev-parse", "--short", "HEAD"],
            cwd=os.path.dirname(os.path.abspath(__file__)),
            capture_output=True,
            text=True,
            timeout=5,
        )
        return out.stdout.strip() if out.returncode == 0 else ""
    except Exception:
        return ""


def version_string() -> str:
    commit = git_commit()
    return f"{__version__}+{commit}" if commit else __version__
