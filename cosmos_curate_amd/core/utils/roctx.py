"""roctx range annotations — the rocprof-visible counterpart of the
reference's @nvtx.annotate on every hot stage (SURVEY.md §2b row 15;
call sites like clip_frame_extraction_stages.py:167,
nvcodec_utils.py:100).  No-op when libroctx64 is absent (dev container).
"""

from __future__ import annotations

import ctypes
import functools
from contextlib import contextmanager

_lib: ctypes.CDLL | None = None
_tried = False


def _load() -> ctypes.CDLL | None:
    global _lib, _tried
    if not _tried:
        _tried = True
        for name in ("libroctx64.so", "libroctx64.so.4", "libroctx64.so.1"):
            try:
                _lib = ctypes.CDLL(name)
                _lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
                break
            except OSError:
                continue
    return _lib


@contextmanager
def roctx_range(name: str):
    lib = _load()
    if lib is not None:
        lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        if lib is not None:
            lib.roctxRangePop()


def annotate(name: str | None = None):
    """Decorator flavor (the reference's @nvtx.annotate shape)."""

    def deco(fn):
        label = name or fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            with roctx_range(label):
                return fn(*args, **kwargs)

        return wrapper

    return deco
