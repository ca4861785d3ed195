"""Demuxer robustness fuzzer (hang/OOB hunting on corrupt containers).

The 120-case bounded version runs in the CPU suite
(tests/test_demux_abi.py::test_demux_corrupt_input_never_crashes); this
standalone tool runs N cases across h264/hevc/multichunk bases including
the remux path.  Round-1 findings it drove (all fixed, cc_demux.cpp):
an unbounded stts/ctts run expansion (hang), stsd entry sizes escaping
the box (OOB read / stall), sample ranges with overflowing offsets, and
inconsistent table lengths reaching the PTS loop.
"""
import argparse
import pathlib
import sys

import numpy as np

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from cosmos_curate_amd import hotpath  # noqa: E402
from oracle import mp4_write  # noqa: E402


def main(cases: int, seed: int) -> None:
    hotpath.load()
    rng = np.random.default_rng(seed)
    bases = [
        mp4_write.write_mp4([40] * 12, stts=[(12, 512)],
                            ctts=[(1, 512 * o) for o in [2, 4, 1, 1] * 3],
                            timescale=12288, elst_media_time=1024,
                            sync_samples=[1, 5, 9]),
        mp4_write.write_mp4([30] * 10, stts=[(10, 400)], ctts=None,
                            timescale=90000, sync_samples=[1],
                            samples_per_chunk=3, use_co64=True),
        mp4_write.write_mp4([25] * 8, stts=[(8, 512)], ctts=None,
                            timescale=12288, sync_samples=[1], codec="hevc"),
    ]
    for i in range(cases):
        base = bases[int(rng.integers(0, len(bases)))]
        buf = bytearray(base)
        mode = rng.integers(0, 3)
        if mode == 0:
            buf = buf[: int(rng.integers(4, len(buf)))]
        elif mode == 1:
            for _ in range(int(rng.integers(1, 10))):
                buf[int(rng.integers(0, len(buf)))] = int(rng.integers(0, 256))
        else:
            off = int(rng.integers(0, max(1, len(buf) - 8)))
            buf[off:off + 4] = int(rng.integers(0, 2 ** 32)).to_bytes(4, "big")
        try:
            with hotpath.Demuxer(bytes(buf)) as d:
                ts = d.timestamps()
                if len(ts):
                    d.packet(0)
                    d.packet(len(ts) - 1)
                try:
                    d.remux_clip(0.0, 1.0)
                except RuntimeError:
                    pass
        except RuntimeError:
            pass
    print(f"{cases} fuzz cases OK")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--cases", type=int, default=5000)
    ap.add_argument("--seed", type=int, default=7)
    a = ap.parse_args()
    main(a.cases, a.seed)
