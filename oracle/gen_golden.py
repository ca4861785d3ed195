"""Generate the golden fixtures under tests/golden/.

Run from the repo root IN THE DEV CONTAINER (needs /root/reference for the
Sintel-derived vectors; the synthetic fixtures regenerate anywhere):

    python -m oracle.gen_golden

Outputs (all committed; /root/reference is never read at test run time):
- sintel_pts.npz        sorted f32 PTS arrays of the reference's two video
                        fixtures (test_clip_10s.mp4, test_video_30s.mp4,
                        Sintel CC-BY), as get_video_timestamps returns them
                        (decoder_utils.py:230-278).
- sintel_meta.json      codec/timescale/dims/frame counts/sync-sample count
                        and sha256 of each fixture's first AnnexB packet.
- synth_bframes.mp4     synthetic MP4 (fake payloads) with B-frame-style
                        ctts + elst shift; exercises reordering paths the
                        Sintel fixtures don't.
- synth_bframes.json    expected sorted PTS for it.
- sampling_kats.json    the reference's own KAT tables for
                        find_closest_indices / sample_closest
                        (test_decoder_utils.py:40-201), as data.
- spans_kats.json       fixed-stride span/UUID expectations
                        (test_fixed_stride_extraction.py:100-320 golden
                        values; UUIDs recomputed per the :554-565 formula).
"""

from __future__ import annotations

import hashlib
import json
import pathlib

import numpy as np

from oracle import mp4_demux, mp4_write, spans

GOLDEN = pathlib.Path(__file__).resolve().parent.parent / "tests" / "golden"
REF_DATA = pathlib.Path("/root/reference/tests/cosmos_curate/pipelines/video/data")


def gen_sintel() -> None:
    if not REF_DATA.exists():
        print("reference fixtures not present; skipping sintel goldens")
        return
    pts = {}
    meta = {}
    for name in ["test_clip_10s.mp4", "test_video_30s.mp4"]:
        data = (REF_DATA / name).read_bytes()
        trk = mp4_demux.parse_mp4(data)[0]
        key = name.replace(".mp4", "")
        pts[key] = trk.pts_seconds_sorted()
        pkts = mp4_demux.annexb_packets(data, trk)
        meta[key] = {
            "codec": trk.codec,
            "timescale": trk.timescale,
            "width": trk.width,
            "height": trk.height,
            "num_samples": len(trk.dts),
            "num_sync": len(trk.sync_samples),
            "elst_media_time": trk.elst_media_time,
            "packet0_sha256": hashlib.sha256(pkts[0]).hexdigest(),
            "packet0_len": len(pkts[0]),
            "packet_lens_first16": [len(p) for p in pkts[:16]],
        }
    np.savez(GOLDEN / "sintel_pts.npz", **pts)
    (GOLDEN / "sintel_meta.json").write_text(json.dumps(meta, indent=1))
    print("wrote sintel goldens")


def gen_synth() -> None:
    # 24 samples at 30 fps (timescale 15360, delta 512), x264-style ctts:
    # IBBP pattern => pts offsets {2,0,1}*512 style, elst media_time = 1024
    # (the decode delay), two keyframes.
    n = 24
    ts = 15360
    delta = 512
    ctts: list[tuple[int, int]] = []
    # x264-style 2-B-frame pattern: decode order [d, d+3, d+1, d+2] with a
    # constant 2-frame pts delay => per-sample ctts offsets [2,4,1,1]*delta,
    # removed at presentation time by elst media_time = 2*delta.
    offsets = [2 * delta, 4 * delta, delta, delta] * (n // 4)
    for off in offsets:
        ctts.append((1, off))
    sizes = [100 + 7 * i for i in range(n)]
    data = mp4_write.write_mp4(
        sizes,
        stts=[(n, delta)],
        ctts=ctts,
        timescale=ts,
        elst_media_time=1024,
        sync_samples=[1, 13],
    )
    (GOLDEN / "synth_bframes.mp4").write_bytes(data)
    # expected: pts = dts + ctts - elst, sorted, /ts, f32
    dts = np.arange(n) * delta
    pts = np.sort(dts + np.array(offsets) - 1024).astype(np.float64) / ts
    expect = [float(np.float32(x)) for x in pts]
    (GOLDEN / "synth_bframes.json").write_text(
        json.dumps({"timescale": ts, "n": n, "pts_sorted": expect}, indent=1)
    )
    print("wrote synth_bframes fixture")


def gen_sampling_kats() -> None:
    """KAT tables from /root/reference/tests/.../test_decoder_utils.py:40-201."""
    kats = {
        "find_closest_indices": [
            {"src": [0, 1, 2, 3, 4], "dst": [0, 1, 2, 3, 4], "expected": [0, 1, 2, 3, 4]},
            {"src": [0, 1, 2, 3, 4], "dst": [0.5, 1.5, 2.5, 3.5, 4.5], "expected": [0, 1, 2, 3, 4]},
            {"src": [0, 1, 2, 3, 4], "dst": [-0.5, 0.5, 1.5, 2.5, 3.5], "expected": [0, 0, 1, 2, 3]},
            {"src": [0, 1, 2, 3, 4], "dst": [0.6, 2.6, 5.6], "expected": [1, 3, 4]},
        ],
        "sample_closest": [
            {"src": list(range(5)), "rate": 1.0, "start": None, "stop": None,
             "endpoint": True, "dedup": True,
             "indices": [0, 1, 2, 3, 4], "counts": [1, 1, 1, 1, 1]},
            {"src": [i / 30.0 for i in range(10)], "rate": 10.0, "start": None,
             "stop": None, "endpoint": True, "dedup": True,
             "indices": [0, 3, 6, 9], "counts": [1, 1, 1, 1]},
            {"src": [i / 30.0 for i in range(10)], "rate": 10.0, "start": 0.1,
             "stop": 0.2, "endpoint": True, "dedup": True,
             "indices": [3, 6], "counts": [1, 1]},
            {"src": [i / 30.0 for i in range(10)], "rate": 10.0, "start": 0.1,
             "stop": 0.2, "endpoint": False, "dedup": True,
             "indices": [3], "counts": [1]},
            {"src": [0.0, 0.1, 0.2, 0.4, 0.5, 0.6], "rate": 5.0, "start": None,
             "stop": None, "endpoint": True, "dedup": True,
             "indices": [0, 2, 3, 5], "counts": [1, 1, 1, 1]},
            {"src": list(range(10)), "rate": 2.0, "start": None, "stop": None,
             "endpoint": False, "dedup": True,
             "indices": [0, 1, 2, 3, 4, 5, 6, 7, 8], "counts": [2] * 9},
            {"src": list(range(10)), "rate": 2.0, "start": None, "stop": None,
             "endpoint": False, "dedup": False,
             "indices": [0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 5, 5, 6, 6, 7, 7, 8, 8],
             "counts": [1] * 18},
            {"src": list(range(10)), "rate": 2.0, "start": None, "stop": None,
             "endpoint": True, "dedup": False,
             "indices": [0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 5, 5, 6, 6, 7, 7, 8, 8, 9],
             "counts": [1] * 19},
        ],
    }
    (GOLDEN / "sampling_kats.json").write_text(json.dumps(kats, indent=1))
    print("wrote sampling KATs")


def gen_spans_kats() -> None:
    """Golden spans from test_fixed_stride_extraction.py + recomputed UUIDs."""
    cases = [
        {"start": 0.0, "end": 30.0, "len": 10.0, "stride": 10.0, "min": 10.0,
         "spans": [[0.0, 10.0], [10.0, 20.0], [20.0, 30.0]]},
        {"start": 0.0, "end": 30.0, "len": 5.0, "stride": 5.0, "min": 5.0,
         "spans": [[0.0, 5.0], [5.0, 10.0], [10.0, 15.0], [15.0, 20.0],
                    [20.0, 25.0], [25.0, 30.0]]},
        # overlap: 10 s clips, 5 s stride, min 2 -> last short span kept
        {"start": 0.0, "end": 30.0, "len": 10.0, "stride": 5.0, "min": 2.0,
         "spans": [[0.0, 10.0], [5.0, 15.0], [10.0, 20.0], [15.0, 25.0],
                    [20.0, 30.0], [25.0, 30.0]]},
        # min-length filter drops trailing 2 s remainder
        {"start": 0.0, "end": 32.0, "len": 10.0, "stride": 10.0, "min": 5.0,
         "spans": [[0.0, 10.0], [10.0, 20.0], [20.0, 30.0]]},
    ]
    out = []
    for c in cases:
        got = spans.make_spans_fixed_stride(c["start"], c["end"], c["len"], c["stride"], c["min"])
        assert got == [tuple(s) for s in c["spans"]], (c, got)
        uuids = spans.make_clip_uuids("session-abc", got)
        out.append({**c, "session_id": "session-abc", "uuids": [str(u) for u in uuids]})
    (GOLDEN / "spans_kats.json").write_text(json.dumps(out, indent=1))
    print("wrote spans KATs")


if __name__ == "__main__":
    GOLDEN.mkdir(parents=True, exist_ok=True)
    gen_sintel()
    gen_synth()
    gen_sampling_kats()
    gen_spans_kats()
