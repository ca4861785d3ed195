"""Oracle span math + UUIDs vs golden expectations.

Golden data: tests/golden/spans_kats.json (fixed-stride expectations of
/root/reference/tests/.../test_fixed_stride_extraction.py:100-320 plus the
uuid5 formula of clip_extraction_stages.py:554-565).
"""

import json
import uuid

from oracle.spans import make_clip_uuids, make_spans_fixed_stride


def test_spans_and_uuids_kats(golden_dir):
    for case in json.loads((golden_dir / "spans_kats.json").read_text()):
        got = make_spans_fixed_stride(
            case["start"], case["end"], case["len"], case["stride"], case["min"]
        )
        assert got == [tuple(s) for s in case["spans"]]
        got_uuids = make_clip_uuids(case["session_id"], got)
        assert [str(u) for u in got_uuids] == case["uuids"]


def test_uuid_formula_is_float_repr():
    """uuid5(NAMESPACE_URL, f"{session}_{s}_{e}") with float repr endpoints."""
    u = make_clip_uuids("sess", [(0.0, 10.0)])[0]
    assert u == uuid.uuid5(uuid.NAMESPACE_URL, "sess_0.0_10.0")


def test_empty_and_short_videos():
    assert make_spans_fixed_stride(0.0, 0.0, 10.0, 10.0, 10.0) == []
    # 7 s video, min 10 s -> no clips
    assert make_spans_fixed_stride(0.0, 7.0, 10.0, 10.0, 10.0) == []
    # 7 s video, min 2 s -> one short clip
    assert make_spans_fixed_stride(0.0, 7.0, 10.0, 10.0, 2.0) == [(0.0, 7.0)]
