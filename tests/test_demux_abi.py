"""C-ABI surface + C++ demuxer parity vs the oracle parser (no GPU).

- every symbol include/cc_hotpath.h declares must be exported by
  libcchot.so (tier contract: the CPU suite checks the ABI loads);
- cc_demux_* output must match oracle/mp4_demux.py bit-for-bit on the
  committed fixtures and on randomized writer-generated containers.
"""

import json
import re
import pathlib

import numpy as np
import pytest

from cosmos_curate_amd import build as cc_build
from cosmos_curate_amd import hotpath
from oracle import mp4_demux, mp4_write

REPO = pathlib.Path(__file__).resolve().parent.parent


@pytest.fixture(scope="session", autouse=True)
def built_lib():
    cc_build.build(verbose=False)
    return hotpath.load()


def test_header_symbols_all_exported(built_lib):
    header = (REPO / "include" / "cc_hotpath.h").read_text()
    declared = set(re.findall(r"\b(cc_[a-z0-9_]+)\s*\(", header))
    lib = built_lib
    missing = [s for s in sorted(declared) if not hasattr(lib, s)]
    assert not missing, f"symbols declared but not exported: {missing}"
    assert len(declared) >= 20


def test_hip_unavailable_is_loud(built_lib):
    """In this GPU-less container the device gate must raise, not fall back."""
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by -m gpu tests")
    with pytest.raises(hotpath.HotpathUnavailableError):
        hotpath.require_gpu()


def test_demux_parity_sintel_goldens(built_lib, golden_dir):
    pts_golden = np.load(golden_dir / "sintel_pts.npz")
    meta = json.loads((golden_dir / "sintel_meta.json").read_text())
    ref_dir = pathlib.Path("/root/reference/tests/cosmos_curate/pipelines/video/data")
    for key in ["test_clip_10s", "test_video_30s"]:
        # On the GPU box /root/reference is absent: fall back to checking the
        # committed golden PTS against the synth fixture only.
        src = ref_dir / f"{key}.mp4"
        if not src.exists():
            pytest.skip("reference fixtures not present (GPU box)")
        data = src.read_bytes()
        with hotpath.Demuxer(data) as d:
            info = d.probe()
            got = d.timestamps()
            pkt0, _, kf0 = d.packet(0)
        assert info.num_samples == meta[key]["num_samples"]
        assert info.width == meta[key]["width"]
        assert info.timescale == meta[key]["timescale"]
        np.testing.assert_array_equal(got, pts_golden[key])
        assert kf0
        import hashlib

        assert hashlib.sha256(pkt0).hexdigest() == meta[key]["packet0_sha256"]
        assert len(pkt0) == meta[key]["packet0_len"]


def test_demux_parity_synth_fixture(built_lib, golden_dir):
    data = (golden_dir / "synth_bframes.mp4").read_bytes()
    exp = json.loads((golden_dir / "synth_bframes.json").read_text())
    with hotpath.Demuxer(data) as d:
        got = d.timestamps()
    np.testing.assert_array_equal(got, np.array(exp["pts_sorted"], dtype=np.float32))


def test_demux_parity_randomized_writers(built_lib):
    """Property test: C++ demuxer == oracle parser on random stts/ctts/elst."""
    rng = np.random.default_rng(0x5EED)
    for _ in range(20):
        n = int(rng.integers(4, 60))
        ts = int(rng.choice([12288, 15360, 90000, 1000]))
        delta = int(rng.integers(100, 4000))
        use_b = bool(rng.integers(0, 2)) and n % 4 == 0
        if use_b:
            ctts = [(1, int(o) * delta) for o in ([2, 4, 1, 1] * (n // 4))]
            elst = 2 * delta
        else:
            ctts, elst = None, None
        sizes = [int(rng.integers(8, 400)) for _ in range(n)]
        data = mp4_write.write_mp4(
            sizes, stts=[(n, delta)], ctts=ctts, timescale=ts,
            elst_media_time=elst, sync_samples=[1],
        )
        oracle_ts = mp4_demux.get_video_timestamps(data)
        with hotpath.Demuxer(data) as d:
            got = d.timestamps()
        np.testing.assert_array_equal(got, oracle_ts)
        # packets match the oracle's AnnexB conversion
        trk = mp4_demux.parse_mp4(data)[0]
        oracle_pkts = mp4_demux.annexb_packets(data, trk)
        with hotpath.Demuxer(data) as d:
            for i in [0, n // 2, n - 1]:
                pkt, _, _ = d.packet(i)
                assert pkt == oracle_pkts[i], f"packet {i} mismatch"


def test_demux_rejects_garbage(built_lib):
    with pytest.raises(RuntimeError):
        hotpath.Demuxer(b"not an mp4 file at all" * 10)


def test_product_metadata_from_demux(built_lib, golden_dir):
    from cosmos_curate_amd.pipelines.video.utils.decoder_utils import (
        extract_video_metadata,
        get_video_timestamps,
    )

    data = (golden_dir / "synth_bframes.mp4").read_bytes()
    md = extract_video_metadata(data)
    assert md.width == 64 and md.height == 64
    assert md.video_codec == "h264"
    assert 29.0 < md.fps < 31.0
    ts = get_video_timestamps(data)
    assert len(ts) == 24


def test_demux_parity_multi_entry_stts(built_lib):
    """VFR-style multi-entry stts (+ large elst offsets): C++ == oracle."""
    rng = np.random.default_rng(0xBEEF)
    for trial in range(12):
        ts = int(rng.choice([12288, 90000, 1000, 600]))
        entries = []
        n = 0
        for _ in range(int(rng.integers(2, 6))):
            cnt = int(rng.integers(1, 12))
            entries.append((cnt, int(rng.integers(50, 5000))))
            n += cnt
        # edit-list media_time several sample durations in (beyond one GOP)
        elst = int(entries[0][1] * rng.integers(0, 5)) if trial % 2 else None
        sizes = [int(rng.integers(8, 200)) for _ in range(n)]
        data = mp4_write.write_mp4(
            sizes, stts=entries, ctts=None, timescale=ts,
            elst_media_time=elst, sync_samples=[1],
        )
        oracle_ts = mp4_demux.get_video_timestamps(data)
        with hotpath.Demuxer(data) as d:
            got = d.timestamps()
            info = d.probe()
        np.testing.assert_array_equal(got, oracle_ts)
        assert info.num_samples == n


def test_demux_parity_hevc_synthetic(built_lib):
    """HEVC (hvc1/hvcC) containers: codec probe, PTS, and VPS/SPS/PPS
    AnnexB prefixing — C++ demuxer == oracle (config #4's 4K HEVC
    container path; decode itself stays behind the rocDecode seam)."""
    rng = np.random.default_rng(0x4EC)
    for _ in range(8):
        n = int(rng.integers(3, 40))
        delta = int(rng.integers(100, 4000))
        sizes = [int(rng.integers(8, 200)) for _ in range(n)]
        data = mp4_write.write_mp4(
            sizes, stts=[(n, delta)], ctts=None, timescale=90000,
            sync_samples=[1], codec="hevc",
        )
        trk = mp4_demux.parse_mp4(data)[0]
        assert trk.codec in ("hvc1", "hev1") and trk.hvcc
        oracle_ts = mp4_demux.get_video_timestamps(data)
        oracle_pkts = mp4_demux.annexb_packets(data, trk)
        with hotpath.Demuxer(data) as d:
            info = d.probe()
            assert info.codec == 1  # hevc
            got = d.timestamps()
            np.testing.assert_array_equal(got, oracle_ts)
            for i in [0, n // 2, n - 1]:
                pkt, _, kf = d.packet(i)
                assert pkt == oracle_pkts[i], f"hevc packet {i} mismatch"
            assert kf is not None


def test_remux_hevc_clip_roundtrip(built_lib):
    """HEVC remux: span [start,end) from a sync sample -> standalone
    hvc1 mp4 that re-demuxes with matching PTS and packets."""
    n, delta, ts = 24, 512, 12288
    sizes = [int(50 + i) for i in range(n)]
    data = mp4_write.write_mp4(
        sizes, stts=[(n, delta)], ctts=None, timescale=ts,
        sync_samples=[1, 13], codec="hevc",
    )
    with hotpath.Demuxer(data) as d:
        clip = d.remux_clip(0.0, 13 * 512 / ts)  # frames 0..12
    with hotpath.Demuxer(clip) as d2:
        info = d2.probe()
        assert info.codec == 1
        assert info.num_samples == 13
        pkt, _, kf = d2.packet(0)
        assert kf
    trk = mp4_demux.parse_mp4(clip)[0]
    assert trk.codec in ("hvc1",) and trk.hvcc


def test_demux_parity_multichunk_co64(built_lib):
    """Multi-chunk stsc layouts and 64-bit chunk offsets (co64): the
    C++ demuxer's per-sample byte ranges == the oracle's."""
    rng = np.random.default_rng(0xC064)
    for spc, co64 in [(1, False), (3, False), (5, True), (2, True)]:
        n = int(rng.integers(6, 30))
        sizes = [int(rng.integers(8, 120)) for _ in range(n)]
        data = mp4_write.write_mp4(
            sizes, stts=[(n, 512)], ctts=None, timescale=12288,
            sync_samples=[1], samples_per_chunk=spc, use_co64=co64,
        )
        trk = mp4_demux.parse_mp4(data)[0]
        oracle_pkts = mp4_demux.annexb_packets(data, trk)
        oracle_ts = mp4_demux.get_video_timestamps(data)
        with hotpath.Demuxer(data) as d:
            np.testing.assert_array_equal(d.timestamps(), oracle_ts)
            for i in [0, n // 2, n - 1]:
                pkt, _, _ = d.packet(i)
                assert pkt == oracle_pkts[i], f"spc={spc} co64={co64} pkt {i}"


def test_demux_corrupt_input_never_crashes(built_lib):
    """Robustness fuzz: truncations and byte-flips of valid containers
    must produce a clean error or a successful parse — never a crash
    (a ctypes-level fault would take the process down)."""
    rng = np.random.default_rng(0xF022)
    base = mp4_write.write_mp4(
        [40] * 12, stts=[(12, 512)], ctts=[(1, 512 * o) for o in [2, 4, 1, 1] * 3],
        timescale=12288, elst_media_time=1024, sync_samples=[1, 5, 9],
    )
    cases = 0
    for _ in range(120):
        buf = bytearray(base)
        mode = rng.integers(0, 3)
        if mode == 0:  # truncate
            buf = buf[: int(rng.integers(4, len(buf)))]
        elif mode == 1:  # flip random bytes
            for _ in range(int(rng.integers(1, 8))):
                buf[int(rng.integers(0, len(buf)))] = int(rng.integers(0, 256))
        else:  # corrupt a box size field
            off = int(rng.integers(0, max(1, len(buf) - 8)))
            buf[off:off + 4] = int(rng.integers(0, 2**32)).to_bytes(4, "big")
        data = bytes(buf)
        try:
            with hotpath.Demuxer(data) as d:
                ts = d.timestamps()
                if len(ts):
                    d.packet(0)
        except RuntimeError:
            pass  # clean error path
        cases += 1
    assert cases == 120
