"""Oracle sampling math vs the reference's own KAT tables.

Golden data: tests/golden/sampling_kats.json, carrying the tables of
/root/reference/tests/cosmos_curate/pipelines/video/utils/
test_decoder_utils.py:40-201 as data (see oracle/gen_golden.py).
"""

import json

import numpy as np
import pytest

from oracle.sampling import broadcast_selected, find_closest_indices, sample_closest


@pytest.fixture(scope="module")
def kats(golden_dir):
    return json.loads((golden_dir / "sampling_kats.json").read_text())


def test_find_closest_indices_kats(kats):
    for case in kats["find_closest_indices"]:
        got = find_closest_indices(
            np.array(case["src"], dtype=np.float32),
            np.array(case["dst"], dtype=np.float32),
        )
        np.testing.assert_array_equal(got, np.array(case["expected"], dtype=np.int32))


def test_sample_closest_kats(kats):
    for case in kats["sample_closest"]:
        idx, counts, _ = sample_closest(
            np.array(case["src"], dtype=np.float32),
            case["rate"],
            start=case["start"],
            stop=case["stop"],
            endpoint=case["endpoint"],
            dedup=case["dedup"],
        )
        np.testing.assert_array_equal(idx, np.array(case["indices"], dtype=np.int32), err_msg=str(case))
        np.testing.assert_array_equal(counts, np.array(case["counts"], dtype=np.int32), err_msg=str(case))


def test_sample_closest_rejects_bad_rate():
    with pytest.raises(ValueError):
        sample_closest(np.arange(5, dtype=np.float32), 0.0)


def test_sample_closest_2fps_10s_clip_is_21_frames():
    """BASELINE workload shape: 10 s 30 fps clip sampled at 2 fps.

    The endpoint-epsilon rule (decoder_utils.py:364-371) includes the final
    frame: grid 0.0..10.0 step 0.5 -> 21 samples, last lands on frame 299.
    """
    ts = (np.arange(300) / 30.0).astype(np.float32)
    idx, counts, _ = sample_closest(ts, 2.0)
    assert counts.sum() == 21
    assert idx[0] == 0 and idx[-1] == 299
    assert np.all(np.diff(idx[:-1]) == 15)


def test_broadcast_selected_duplicates():
    frames = np.arange(4 * 2 * 2 * 3, dtype=np.uint8).reshape(4, 2, 2, 3)
    out = broadcast_selected(frames, np.array([0, 2], dtype=np.int32), np.array([2, 3], dtype=np.int32))
    assert out.shape[0] == 5
    assert np.array_equal(out[0], out[1]) and np.array_equal(out[0], frames[0])
    assert np.array_equal(out[2], frames[2]) and np.array_equal(out[4], frames[2])
