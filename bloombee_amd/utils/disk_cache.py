"""Disk-cache management (parity: reference utils/disk_cache.py —
`allow_cache_reads/writes` locks and `free_disk_space_for` eviction of
least-recently-used cached weight dirs)."""
from __future__ import annotations

import contextlib
import os
import shutil
import threading
from pathlib import Path
from typing import Optional

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

DEFAULT_CACHE_DIR = os.path.expanduser(
    os.environ.get("BBAMD_CACHE_DIR", "~/.cache/bloombee_amd"))

_lock = threading.Lock()


@contextlib.contextmanager
def allow_cache_reads(cache_dir: Optional[str] = None):
    with _lock:
        yield Path(cache_dir or DEFAULT_CACHE_DIR)


@contextlib.contextmanager
def allow_cache_writes(cache_dir: Optional[str] = None):
    with _lock:
        d = Path(cache_dir or DEFAULT_CACHE_DIR)
        d.mkdir(parents=True, exist_ok=True)
        yield d


def free_disk_space_for(size_bytes: int, cache_dir: Optional[str] = None) -> bool:
    """Evict least-recently-used top-level cache entries until `size_bytes`
    fits in the filesystem's free space. Returns True if enough is free."""
    d = Path(cache_dir or DEFAULT_CACHE_DIR)
    if not d.exists():
        return True

    def free() -> int:
        st = os.statvfs(str(d))
        return st.f_bavail * st.f_frsize

    if free() >= size_bytes:
        return True
    entries = sorted(d.iterdir(), key=lambda p: p.stat().st_atime)
    for e in entries:
        if free() >= size_bytes:
            return True
        logger.info("evicting cached entry %s to free disk space", e)
        shutil.rmtree(e, ignore_errors=True) if e.is_dir() else e.unlink(missing_ok=True)
    return free() >= size_bytes
