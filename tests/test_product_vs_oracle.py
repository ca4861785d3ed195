"""Product host-side math vs oracle (CPU): sampling, spans, UUIDs, signatures."""

import json

import numpy as np
import pytest

from cosmos_curate_amd.pipelines.video.clipping.clip_extraction_stages import (
    _make_clip_uuids,
    _make_spans_fixed_stride,
)
from cosmos_curate_amd.pipelines.video.utils.decoder_utils import (
    FrameExtractionPolicy,
    FrameExtractionSignature,
    find_closest_indices,
    sample_closest,
)
from oracle import sampling as osampling
from oracle import spans as ospans


def test_sampling_matches_oracle_randomized():
    rng = np.random.default_rng(7)
    for _ in range(50):
        n = int(rng.integers(2, 400))
        deltas = rng.uniform(0.01, 0.1, size=n).astype(np.float32)
        src = np.cumsum(deltas).astype(np.float32)
        rate = float(rng.uniform(0.3, 40.0))
        endpoint = bool(rng.integers(0, 2))
        dedup = bool(rng.integers(0, 2))
        got = sample_closest(src, rate, endpoint=endpoint, dedup=dedup)
        want = osampling.sample_closest(src, rate, endpoint=endpoint, dedup=dedup)
        np.testing.assert_array_equal(got[0], want[0])
        np.testing.assert_array_equal(got[1], want[1])

        dst = np.sort(rng.uniform(-1, src[-1] + 1, size=17).astype(np.float32))
        np.testing.assert_array_equal(
            find_closest_indices(src, dst), osampling.find_closest_indices(src, dst)
        )


def test_sampling_kats(golden_dir):
    kats = json.loads((golden_dir / "sampling_kats.json").read_text())
    for case in kats["sample_closest"]:
        idx, counts, _ = sample_closest(
            np.array(case["src"], dtype=np.float32),
            case["rate"],
            start=case["start"],
            stop=case["stop"],
            endpoint=case["endpoint"],
            dedup=case["dedup"],
        )
        np.testing.assert_array_equal(idx, np.array(case["indices"], dtype=np.int32))
        np.testing.assert_array_equal(counts, np.array(case["counts"], dtype=np.int32))


def test_spans_and_uuids_match_oracle(golden_dir):
    for case in json.loads((golden_dir / "spans_kats.json").read_text()):
        got = _make_spans_fixed_stride(
            case["start"], case["end"], case["len"], case["stride"], case["min"]
        )
        want = ospans.make_spans_fixed_stride(
            case["start"], case["end"], case["len"], case["stride"], case["min"]
        )
        assert got == want == [tuple(s) for s in case["spans"]]
        assert [str(u) for u in _make_clip_uuids(case["session_id"], got)] == case["uuids"]
        assert _make_clip_uuids(case["session_id"], got) == ospans.make_clip_uuids(
            case["session_id"], got
        )


def test_signature_format_matches_reference():
    """`"{policy!s}-{int(fps*1000)}"` (decoder_utils.py:110-117) — e.g.
    'FrameExtractionPolicy.sequence-2000'."""
    sig = FrameExtractionSignature(FrameExtractionPolicy.sequence, 2.0)
    assert sig.to_str() == "FrameExtractionPolicy.sequence-2000"
    sig = FrameExtractionSignature(FrameExtractionPolicy.middle, 0.5)
    assert sig.to_str() == "FrameExtractionPolicy.middle-500"


def test_bad_sample_rate_raises():
    with pytest.raises(ValueError):
        sample_closest(np.arange(4, dtype=np.float32), -1.0)


def test_sample_closest_extreme_ratios():
    """find_closest/sample_closest at extreme fps ratios: product ==
    oracle for upsampling (target >> source), near-equal rates, and
    very sparse sampling (decoder_utils.py:356-453 semantics)."""
    import numpy as np

    from cosmos_curate_amd.pipelines.video.utils import decoder_utils as du
    from oracle import sampling as osamp

    rng = np.random.default_rng(0xA7E5)
    cases = []
    for _ in range(40):
        fps = float(rng.choice([10, 23.976, 24, 29.97, 30, 59.94, 120]))
        n = int(rng.integers(2, 400))
        target = float(rng.choice([0.1, 0.5, 1, 2, 5, fps, fps * 2, 240]))
        cases.append((fps, n, target))
    cases += [(30.0, 1, 2.0), (30.0, 2, 240.0), (120.0, 300, 0.1)]
    for fps, n, target in cases:
        ts = (np.arange(n) / fps).astype(np.float32)
        gi, gc, gs = du.sample_closest(ts, sample_rate=target)
        oi, oc, os_ = osamp.sample_closest(ts, target)
        np.testing.assert_array_equal(gi, oi, err_msg=f"{fps=} {n=} {target=}")
        np.testing.assert_array_equal(gc, oc)
        np.testing.assert_array_equal(gs, os_)


def test_fixed_stride_spans_randomized_vs_oracle():
    """Randomized stride/len/min combos: product span math == oracle
    (clip_extraction_stages.py:512-551), including overlap (stride <
    len), sparse (stride > len), and fractional-second configurations."""
    import numpy as np

    from cosmos_curate_amd.pipelines.video.clipping.clip_extraction_stages import (
        _make_clip_uuids,
        _make_spans_fixed_stride,
    )
    from oracle import spans as ospans

    rng = np.random.default_rng(0x5A9)
    for _ in range(60):
        end = float(rng.uniform(0.5, 120.0))
        clip_len = float(rng.uniform(0.5, 30.0))
        stride = float(rng.uniform(0.25, 30.0))
        min_len = float(rng.uniform(0.0, clip_len))
        got = _make_spans_fixed_stride(0.0, end, clip_len, stride, min_len)
        want = ospans.make_spans_fixed_stride(0.0, end, clip_len, stride, min_len)
        assert got == want, (end, clip_len, stride, min_len)
        assert _make_clip_uuids("s", got) == ospans.make_clip_uuids("s", want)
