"""Clip span math + deterministic clip UUIDs (CPU oracle).

Restates /root/reference/cosmos_curate/pipelines/video/clipping/
clip_extraction_stages.py:
- ``make_spans_fixed_stride``  (:512-551)
- ``make_clip_uuids``          (:554-565)  uuid5(NAMESPACE_URL, f"{session}_{s}_{e}")
  with the span endpoints formatted by Python float repr -- the bit-exact
  UUID contract.

Pinned by the reference's fixed-stride golden expectations
(test_fixed_stride_extraction.py:100-320): 30 s video, 10s/10s/10s ->
[(0,10),(10,20),(20,30)]; 5s/5s/5s -> 6 spans; carried as data in
tests/golden/spans_kats.json.
"""

from __future__ import annotations

import uuid


def make_spans_fixed_stride(
    start_s: float,
    end_s: float,
    clip_len_s: float,
    clip_stride_s: float,
    min_clip_length_s: float,
) -> list[tuple[float, float]]:
    """Fixed-stride spans over [start_s, end_s).

    Semantics (clip_extraction_stages.py:512-551): walk start positions by
    clip_stride_s while strictly below end_s; each span ends at
    min(start + clip_len_s, end_s); keep spans of length >= min_clip_length_s.
    Float accumulation order matters for bit-exactness: the walk is a
    repeated ``+= clip_stride_s`` on a Python float, not an arange.
    """
    spans: list[tuple[float, float]] = []
    pos = start_s
    while pos < end_s:
        span_end = min(pos + clip_len_s, end_s)
        if (span_end - pos) >= min_clip_length_s:
            spans.append((pos, span_end))
        pos += clip_stride_s
    return spans


def make_clip_uuids(session_id: str, spans: list[tuple[float, float]]) -> list[uuid.UUID]:
    """Deterministic clip UUIDs (clip_extraction_stages.py:554-565).

    uuid5 over NAMESPACE_URL of "{session_id}_{start}_{end}" where start/end
    render via Python float str() (f-string repr of the tuple elements).
    """
    return [
        uuid.uuid5(uuid.NAMESPACE_URL, f"{session_id}_{s}_{e}") for (s, e) in spans
    ]
