"""glibc malloc tuning for large wire buffers.

Every turbo request/response allocates a fresh multi-MB bytes object; with
glibc defaults those come from mmap and are returned to the kernel on free,
so each request pays ~4700 first-touch page faults per 19 MB buffer. Raising
M_MMAP_THRESHOLD keeps them in the (reused, already-faulted) main arena;
raising M_TRIM_THRESHOLD stops the arena being trimmed back.
"""
from __future__ import annotations

import ctypes
import ctypes.util

_M_TRIM_THRESHOLD = -1
_M_MMAP_THRESHOLD = -3

_done = False


def tune_malloc(threshold: int = 512 << 20) -> bool:
    """Idempotent; returns True if mallopt succeeded."""
    global _done
    if _done:
        return True
    try:
        libc = ctypes.CDLL(ctypes.util.find_library("c") or "libc.so.6",
                           use_errno=True)
        ok1 = libc.mallopt(_M_MMAP_THRESHOLD, threshold)
        ok2 = libc.mallopt(_M_TRIM_THRESHOLD, threshold)
        _done = bool(ok1 and ok2)
    except Exception:
        _done = False
    return _done
