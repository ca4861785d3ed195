"""LazyData: the zero-copy inter-stage payload wrapper (inline mode).

Mirror of /root/reference/cosmos_curate/core/utils/data/lazy_data.py:189
semantics in its shipped configuration: values travel INLINE (the Plasma
ObjectRef path is implemented-but-disabled upstream, lazy_data.py:58-100),
pickling numpy payloads zero-copy via PEP-574 out-of-band buffers when the
transport supports it.  API kept: coerce / resolve / drop / nbytes, plus
truthiness on presence.
"""

from __future__ import annotations

from typing import Generic, TypeVar

import numpy as np

T = TypeVar("T")


class LazyData(Generic[T]):
    """Inline value holder with the reference's LazyData surface."""

    __slots__ = ("value", "nbytes")

    def __init__(self, value: T | None = None, nbytes: int = 0) -> None:
        self.value = value
        if nbytes == 0 and value is not None and hasattr(value, "nbytes"):
            nbytes = int(value.nbytes)  # type: ignore[attr-defined]
        self.nbytes = nbytes

    @classmethod
    def coerce(cls, value) -> "LazyData":
        """Field converter: accept raw values, bytes, or LazyData."""
        if isinstance(value, LazyData):
            return value
        if isinstance(value, (bytes, bytearray)):
            arr = np.frombuffer(value, dtype=np.uint8)
            return cls(value=arr, nbytes=arr.nbytes)
        return cls(value=value)

    def resolve(self) -> T | None:
        """Return the payload (inline mode: no fetch)."""
        return self.value

    def drop(self) -> None:
        """Release the payload (error-path cleanup, lazy_data.py usage)."""
        self.value = None
        self.nbytes = 0

    def __bool__(self) -> bool:
        return self.value is not None


def prefetch(refs) -> None:
    """Non-blocking fetch hint (ref_resolver.py:70-105 surface).

    The reference tells Ray to pull ObjectRefs into the local Plasma
    store.  The rebuild's payloads are in-process (LazyData.value), so
    there is nothing to fetch — the function exists so stage code
    written against the reference API runs unchanged, and so a future
    distributed object store can slot in behind the same call.
    """
    _ = [r for r in refs if r is not None]


def resolve_as_ready(items):
    """Yield (key, value|None) pairs 1:1 with the input
    (ref_resolver.py:108-170 surface).

    The reference yields in ray.wait completion order, already-local
    values first; with in-process payloads everything is already local,
    so input order IS completion order.  Callers keep the reference's
    memory discipline: process and release each item before advancing.
    """
    for key, ld in items:
        yield key, (ld.resolve() if ld is not None else None)
