"""Atomic small-file persistence helpers (checkpoints, CDI specs).

Hot-path notes (the prepare path writes two of these per claim): the
original implementation paid ``tempfile.mkstemp`` (random-name loop +
O_EXCL probing, ~3% of lifecycle CPU in the sampling profile) and a
``makedirs`` per write. Temp names here are unique by construction
(pid + per-process counter — collisions impossible within a process,
and cross-process names differ by pid), and directory existence is
memoized (directories are created once per process lifetime; an
externally deleted parent resurfaces as ENOENT and retries once).
"""

from __future__ import annotations

import itertools
import json
import os
import threading
from typing import Any

_counter = itertools.count()
_known_dirs = set()
_dirs_lock = threading.Lock()


def _ensure_dir(d: str) -> None:
    if d in _known_dirs:
        return
    os.makedirs(d, exist_ok=True)
    with _dirs_lock:
        _known_dirs.add(d)


def _tmp_name(path: str) -> str:
    d = os.path.dirname(path)
    return os.path.join(d, f".tmp-{os.getpid()}-{next(_counter)}")


def atomic_write_text(path: str, data: str, *, durable: bool = True) -> None:
    """tmp + fsync + rename so a crash can never leave a torn file.

    ``durable=False`` skips the fsync (rename atomicity is kept: readers
    never see a torn file, but the content may be lost on power failure).
    Only correct for files that are REGENERABLE from durable state —
    e.g. per-claim CDI specs, which DeviceState rebuilds from the
    checkpoint on the prepare cache-hit path. fsync is ~10% of the
    prepare hot path on the MI355X pool boxes (sampling profile,
    profiles/round2_hardware_notes.md)."""
    d = os.path.dirname(path)
    _ensure_dir(d)
    tmp = _tmp_name(path)
    try:
        fd = os.open(tmp, os.O_WRONLY | os.O_CREAT | os.O_EXCL, 0o644)
    except FileNotFoundError:
        # parent deleted externally since we memoized it: recreate
        with _dirs_lock:
            _known_dirs.discard(d)
        _ensure_dir(d)
        fd = os.open(tmp, os.O_WRONLY | os.O_CREAT | os.O_EXCL, 0o644)
    try:
        os.write(fd, data.encode())
        if durable:
            os.fsync(fd)
        os.close(fd)
        fd = -1
        os.replace(tmp, path)
    except BaseException:
        if fd >= 0:
            os.close(fd)
        try:
            os.unlink(tmp)
        except OSError:
            pass
        raise


def atomic_write_json(path: str, obj: Any) -> None:
    atomic_write_text(
        path, json.dumps(obj, sort_keys=True, separators=(",", ":"))
    )


def read_json(path: str) -> Any:
    with open(path) as f:
        return json.load(f)
