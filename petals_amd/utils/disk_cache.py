"""Disk LRU cache for local model checkpoints (parity: reference
utils/disk_cache.py — fcntl shared/exclusive locks + LRU eviction to fit
max_disk_space). Offline build: the cache holds locally converted/quantized
block shards rather than HF Hub downloads."""

from __future__ import annotations

import contextlib
import fcntl
import logging
import os
import shutil
import time
from pathlib import Path
from typing import Optional

logger = logging.getLogger(__name__)

DEFAULT_CACHE_DIR = Path(os.environ.get("PETALS_AMD_CACHE", Path.home() / ".cache" / "petals_amd"))
BLOCKS_LOCK_FILE = "blocks.lock"


@contextlib.contextmanager
def _blocks_lock(cache_dir: Optional[Path], mode: int):
    cache_dir = Path(cache_dir or DEFAULT_CACHE_DIR)
    cache_dir.mkdir(parents=True, exist_ok=True)
    lock_path = cache_dir / BLOCKS_LOCK_FILE
    with open(lock_path, "wb") as lock_fd:
        fcntl.flock(lock_fd.fileno(), mode)
        yield


def allow_cache_reads(cache_dir: Optional[Path] = None):
    """Shared lock: multiple processes may read the cache concurrently."""
    return _blocks_lock(cache_dir, fcntl.LOCK_SH)


def allow_cache_writes(cache_dir: Optional[Path] = None):
    """Exclusive lock for writes/evictions."""
    return _blocks_lock(cache_dir, fcntl.LOCK_EX)


def _dir_size(path: Path) -> int:
    total = 0
    for root, _dirs, files in os.walk(path):
        for f in files:
            with contextlib.suppress(OSError):
                total += os.path.getsize(os.path.join(root, f))
    return total


def free_disk_space_for(size_bytes: int, *, cache_dir: Optional[Path] = None, max_disk_space: Optional[int] = None):
    """Evict least-recently-used cache entries until `size_bytes` more fits
    under `max_disk_space` (and under the filesystem's free space)."""
    cache_dir = Path(cache_dir or DEFAULT_CACHE_DIR)
    if not cache_dir.exists():
        return
    entries = [p for p in cache_dir.iterdir() if p.name != BLOCKS_LOCK_FILE]
    entries.sort(key=lambda p: p.stat().st_atime)

    def usage() -> int:
        return _dir_size(cache_dir)

    def available() -> int:
        stat = shutil.disk_usage(cache_dir)
        budget = stat.free
        if max_disk_space is not None:
            budget = min(budget, max_disk_space - usage())
        return budget

    for entry in entries:
        if available() >= size_bytes:
            return
        logger.info("LRU-evicting %s to free disk space", entry)
        if entry.is_dir():
            shutil.rmtree(entry, ignore_errors=True)
        else:
            with contextlib.suppress(OSError):
                entry.unlink()
    if available() < size_bytes:
        logger.warning("could not free %d bytes in %s", size_bytes, cache_dir)


def touch(path: Path) -> None:
    """Mark a cache entry as recently used."""
    with contextlib.suppress(OSError):
        os.utime(path, (time.time(), os.stat(path).st_mtime))
