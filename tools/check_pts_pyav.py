"""PTS cross-check against PyAV — run wherever PyAV exists.

Neither this container nor the GPU boxes ship PyAV/ffmpeg (verified:
DESIGN.md §7), so the demuxer's external pins are (a) the H.264
bitstream POC/SPS checks (tests/test_demux_bitstream_pin.py, run
in-image) and (b) THIS script, committed for any environment that has
``pip install av``: it compares oracle/mp4_demux.py's sorted-PTS
contract (and the committed sintel_pts.npz goldens) against PyAV's
packet timestamps — the exact upstream recipe
(/root/reference .../decoder_utils.py:230-278: pts * time_base as
float32, sorted).

    python tools/check_pts_pyav.py <video.mp4> [...]
    python tools/check_pts_pyav.py --goldens   # verify tests/golden/*.npz

Exit 0 = all files bit-identical; nonzero otherwise.
"""

from __future__ import annotations

import pathlib
import sys

import numpy as np

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

from oracle import mp4_demux  # noqa: E402

GOLDEN = pathlib.Path(__file__).resolve().parent.parent / "tests" / "golden"
REF_DATA = pathlib.Path(
    "/root/reference/tests/cosmos_curate/pipelines/video/data")


def pyav_sorted_pts(path: pathlib.Path) -> np.ndarray:
    import av  # noqa: PLC0415 — only available off-image

    with av.open(str(path)) as container:
        stream = container.streams.video[0]
        tb = stream.time_base
        ts = [
            np.float32(float(p.pts * tb))
            for p in container.demux(stream)
            if p.pts is not None
        ]
    return np.sort(np.array(ts, dtype=np.float32))


def check_one(path: pathlib.Path) -> bool:
    ours = mp4_demux.get_video_timestamps(path.read_bytes())
    ref = pyav_sorted_pts(path)
    ok = ours.shape == ref.shape and np.array_equal(ours, ref)
    status = "OK (bit-identical)" if ok else "MISMATCH"
    print(f"{path.name}: {len(ours)} vs {len(ref)} samples -> {status}")
    if not ok and ours.shape == ref.shape:
        d = np.nonzero(ours != ref)[0]
        print(f"  first diffs at {d[:5]}: ours={ours[d[:5]]} pyav={ref[d[:5]]}")
    return ok


def main(argv: list[str]) -> int:
    bad = 0
    if argv and argv[0] == "--goldens":
        pts = np.load(GOLDEN / "sintel_pts.npz")
        for name in pts.files:
            path = REF_DATA / f"{name}.mp4"
            if not path.is_file():
                print(f"{name}: fixture not present here, skipping")
                continue
            ref = pyav_sorted_pts(path)
            ok = np.array_equal(pts[name], ref)
            print(f"golden {name}: {'OK' if ok else 'MISMATCH'} vs PyAV")
            bad += 0 if ok else 1
        return bad
    for arg in argv:
        bad += 0 if check_one(pathlib.Path(arg)) else 1
    return bad


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
