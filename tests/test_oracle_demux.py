"""MP4 demux oracle vs committed golden vectors.

The sintel_* goldens were derived in the dev container from the reference's
own media fixtures (Sintel, CC-BY) by oracle/gen_golden.py; the synth fixture
is self-contained and regenerable anywhere.
"""

import json

import numpy as np

from oracle import mp4_demux, mp4_write


def test_synth_bframes_pts(golden_dir):
    data = (golden_dir / "synth_bframes.mp4").read_bytes()
    exp = json.loads((golden_dir / "synth_bframes.json").read_text())
    got = mp4_demux.get_video_timestamps(data)
    assert len(got) == exp["n"]
    np.testing.assert_array_equal(got, np.array(exp["pts_sorted"], dtype=np.float32))


def test_sintel_pts_roundtrip(golden_dir):
    """PTS arrays recorded from the reference fixtures stay bit-stable."""
    pts = np.load(golden_dir / "sintel_pts.npz")
    meta = json.loads((golden_dir / "sintel_meta.json").read_text())
    for key in ["test_clip_10s", "test_video_30s"]:
        arr = pts[key]
        m = meta[key]
        assert arr.dtype == np.float32
        assert len(arr) == m["num_samples"]
        assert arr[0] == np.float32(0.0)
        assert np.all(np.diff(arr) > 0)
        # 24 fps content: constant delta 1/24 at timescale 12288 (f32 rounding
        # of k/24 makes successive diffs wobble by ~1.3e-6)
        np.testing.assert_allclose(np.diff(arr), 1.0 / 24.0, atol=3e-6)


def test_writer_parser_roundtrip_variants():
    """Writer->parser property: stts/ctts/elst round-trip across shapes."""
    ts = 90000
    # no B-frames, no elst
    data = mp4_write.write_mp4([10] * 5, stts=[(5, 3000)], ctts=None, timescale=ts)
    got = mp4_demux.get_video_timestamps(data)
    np.testing.assert_array_equal(
        got, (np.arange(5) * 3000 / ts).astype(np.float32)
    )
    # variable frame durations
    data = mp4_write.write_mp4(
        [10] * 4, stts=[(2, 3000), (2, 1500)], ctts=None, timescale=ts
    )
    got = mp4_demux.get_video_timestamps(data)
    np.testing.assert_array_equal(
        got, (np.array([0, 3000, 6000, 7500]) / ts).astype(np.float32)
    )


def test_annexb_packets_synth(golden_dir):
    data = (golden_dir / "synth_bframes.mp4").read_bytes()
    trk = mp4_demux.parse_mp4(data)[0]
    pkts = mp4_demux.annexb_packets(data, trk)
    assert len(pkts) == len(trk.sizes)
    # sync samples 1 and 13 carry the SPS/PPS prefix
    assert pkts[0][:4] == b"\x00\x00\x00\x01"
    assert len(pkts[0]) > len(pkts[1])
