"""Split driver + transcode + writer on CPU (GPU stages disabled).

Exercises build_input_data -> download -> fixed-stride -> transcode
(raw stream-copy + chunk fan-out) -> writer -> summary.json without
embeddings (the GPU-only stages are covered in -m gpu tests).
"""

import argparse
import json

import numpy as np
import pytest

from cosmos_curate_amd.core.interfaces import SequentialRunner
from cosmos_curate_amd.pipelines.video.clipping.clip_extraction_stages import (
    ClipTranscodingStage,
    chunk_tasks,
)
from cosmos_curate_amd.pipelines.video.splitting_pipeline import (
    _setup_parser,
    split,
)
from cosmos_curate_amd.pipelines.video.utils import raw_backend


@pytest.fixture
def corpus(tmp_path):
    """3 synthetic raw-NV12 'videos' of 30 s @ 30 fps, 64x96."""
    inp = tmp_path / "in"
    inp.mkdir()
    for i in range(3):
        raw = raw_backend.make_synthetic_clip(900, 64, 96, 30, seed=i)
        (inp / f"video_{i}.nv12").write_bytes(raw)
    return inp


def parse(argv):
    p = argparse.ArgumentParser()
    _setup_parser(p)
    return p.parse_args(argv)


def test_split_fixed_stride_no_embeddings(corpus, tmp_path):
    out = tmp_path / "out"
    args = parse(
        [
            "--input-video-path", str(corpus),
            "--output-clip-path", str(out),
            "--no-embeddings",
        ]
    )
    summary = split(args, runner=SequentialRunner())
    assert summary["num_input_videos"] == 3
    assert summary["num_clips"] == 9  # 30 s / 10 s x 3 videos
    assert summary["num_clips_with_errors"] == 0
    disk = json.loads((out / "summary.json").read_text())
    assert disk["num_clips"] == 9
    # per-clip payloads + metadata written
    clips = list((out / "clips").glob("*.bin"))
    metas = list((out / "metas" / "v0").glob("*.json"))
    assert len(clips) == 9 and len(metas) == 9
    # each clip payload is a valid 10 s raw-NV12 stream (300 frames)
    raw = clips[0].read_bytes()
    n, h, w, fps = raw_backend.parse_header(raw)
    assert (n, h, w, fps) == (300, 64, 96, 30.0)
    # stage perf recorded for the whole chain
    for stage in ["VideoDownloader", "FixedStrideExtractorStage",
                  "ClipTranscodingStage", "ClipWriterStage"]:
        assert stage in disk["stage_perf"], disk["stage_perf"].keys()


def test_transcode_slices_are_frame_exact(corpus):
    """Clip payload frames == source frames of the span (bit-exact)."""
    import pathlib
    import uuid as uuid_mod

    from cosmos_curate_amd.core.interfaces import run_pipeline
    from cosmos_curate_amd.pipelines.video.utils.data_model import (
        Clip,
        SplitPipeTask,
        Video,
        VideoMetadata,
    )

    src = (corpus / "video_0.nv12").read_bytes()
    v = Video(
        input_video=pathlib.Path(corpus / "video_0.nv12"),
        metadata=VideoMetadata(size=len(src), height=64, width=96, framerate=30.0,
                               num_frames=900, duration=30.0, video_codec="raw"),
        encoded_data=np.frombuffer(src, dtype=np.uint8),
        timestamps=raw_backend.timestamps(src),
    )
    v.clips.append(Clip(uuid=uuid_mod.uuid4(), source_video="s", span=(10.0, 20.0)))
    out = run_pipeline([SplitPipeTask(videos=[v])], [ClipTranscodingStage()],
                       runner=SequentialRunner())
    clip = out[0].video.clips[0]
    assert not clip.errors
    payload = bytes(clip.encoded_data.resolve())
    idx = np.arange(300, 600, dtype=np.int32)
    want_y, want_uv = raw_backend.frame_planes(src, idx)
    got_y, got_uv = raw_backend.frame_planes(payload, np.arange(300, dtype=np.int32))
    np.testing.assert_array_equal(got_y, want_y)
    np.testing.assert_array_equal(got_uv, want_uv)
    # source bytes dropped after transcode (reference behavior: per-clip
    # payloads replace the source)
    assert out[0].video.encoded_data.resolve() is None


def test_chunk_tasks_fan_out(corpus):
    import pathlib
    import uuid as uuid_mod

    from cosmos_curate_amd.pipelines.video.utils.data_model import (
        Clip,
        SplitPipeTask,
        Video,
    )

    v = Video(input_video=pathlib.Path("/x.mp4"))
    for i in range(20):
        v.clips.append(Clip(uuid=uuid_mod.uuid4(), source_video="s", span=(i, i + 1.0)))
    tasks = chunk_tasks([SplitPipeTask(videos=[v])], num_clips_per_chunk=1)
    # 20 clips / (1*8) per chunk -> 3 subtasks of 8/8/4
    assert [len(t.video.clips) for t in tasks] == [8, 8, 4]
    assert [t.video.clip_chunk_index for t in tasks] == [0, 1, 2]
    assert all(t.video.num_clip_chunks == 3 for t in tasks)
    assert all(t.video.num_total_clips == 20 for t in tasks)
    assert {c.uuid for t in tasks for c in t.video.clips} == {c.uuid for c in v.clips}


def test_mp4_transcode_stream_copy_and_sync_guard(golden_dir):
    """H.264 spans stream-copy sample-exact; non-keyframe-aligned spans
    record a loud per-clip error, never silently re-encode."""
    import pathlib
    import uuid as uuid_mod

    from cosmos_curate_amd.core.interfaces import run_pipeline
    from cosmos_curate_amd.pipelines.video.utils.data_model import (
        Clip,
        SplitPipeTask,
        Video,
    )
    from oracle import mp4_demux

    data = (golden_dir / "synth_bframes.mp4").read_bytes()
    v = Video(input_video=pathlib.Path("/synth.mp4"),
              encoded_data=np.frombuffer(data, dtype=np.uint8))
    # sync samples are 1 and 13 (30 fps): span from 0.0 copies; span from
    # 0.2 s (sample 7, mid-GOP) must fail loudly
    v.clips.append(Clip(uuid=uuid_mod.uuid4(), source_video="s", span=(0.0, 0.4)))
    v.clips.append(Clip(uuid=uuid_mod.uuid4(), source_video="s", span=(0.2, 0.4)))
    out = run_pipeline([SplitPipeTask(videos=[v])], [ClipTranscodingStage()],
                       runner=SequentialRunner())
    good, bad = out[0].video.clips
    assert not good.errors
    clip_bytes = bytes(good.encoded_data.resolve())
    trk = mp4_demux.parse_mp4(clip_bytes)[0]
    assert len(trk.dts) == 12  # frames 0..11 display in [0.0, 0.4)
    src = mp4_demux.parse_mp4(data)[0]
    for j in [0, 5, 11]:
        so, ss = src.offsets[j], src.sizes[j]
        co, cs = trk.offsets[j], trk.sizes[j]
        assert cs == ss and clip_bytes[co:co + cs] == data[so:so + ss]
    assert "transcode" in bad.errors and "sync" in bad.errors["transcode"]


def test_mp4_remux_sintel_head_span(golden_dir):
    """Reference fixture (single-IDR): [0,10 s) span copies sample-exact."""
    import pathlib

    from cosmos_curate_amd import hotpath
    from oracle import mp4_demux

    src_path = pathlib.Path(
        "/root/reference/tests/cosmos_curate/pipelines/video/data/test_video_30s.mp4"
    )
    if not src_path.exists():
        pytest.skip("reference fixtures not present (GPU box)")
    data = src_path.read_bytes()
    src = mp4_demux.parse_mp4(data)[0]
    with hotpath.Demuxer(data) as d:
        clip = d.remux_clip(0.0, 10.0)
        with pytest.raises(RuntimeError, match="sync"):
            d.remux_clip(10.0, 20.0)  # only sample 1 is IDR in this encode
    trk = mp4_demux.parse_mp4(clip)[0]
    assert len(trk.dts) == 240
    pts = trk.pts_seconds_sorted()
    assert abs(pts[0]) < 1e-6 and abs(pts[-1] - 239 / 24) < 1e-4
    for j in [0, 100, 239]:
        so, ss = src.offsets[j], src.sizes[j]
        co, cs = trk.offsets[j], trk.sizes[j]
        assert cs == ss and clip[co:co + cs] == data[so:so + ss]
    assert mp4_demux.annexb_packets(clip, trk)[0][:4] == b"\x00\x00\x00\x01"


def test_multicam_driver_input_and_dry_run(tmp_path):
    """--multi-cam groups UUID session dirs into one task (primary camera
    first, video_pipe_input.py:238-283); --dry-run prints the stage list."""
    import argparse
    import uuid as uuid_mod

    from cosmos_curate_amd.pipelines.video.splitting_pipeline import (
        _setup_parser,
        build_input_data,
        split,
    )
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    root = tmp_path / "sessions"
    sid = str(uuid_mod.uuid4())
    for cam in ["rear", "front"]:
        d = root / sid / cam
        d.mkdir(parents=True)
        (d / "v.nv12").write_bytes(
            raw_backend.make_synthetic_clip(60, 32, 48, 30, seed=1))
    (root / "not-a-session").mkdir()  # ignored: not a UUID name

    p = argparse.ArgumentParser()
    _setup_parser(p)
    args = p.parse_args([
        "--input-video-path", str(root), "--output-clip-path",
        str(tmp_path / "out"), "--multi-cam",
        "--primary-camera-keyword", "front", "--dry-run",
    ])
    tasks = build_input_data(args)
    assert len(tasks) == 1 and tasks[0].session_id == sid
    assert len(tasks[0].videos) == 2
    assert "front" in str(tasks[0].videos[0].input_video)  # primary first

    summary = split(args)
    assert summary["dry_run"] and summary["num_input_videos"] == 1

    # transnetv2 + multicam is rejected loudly (reference :213)
    args2 = p.parse_args([
        "--input-video-path", str(root), "--output-clip-path",
        str(tmp_path / "o2"), "--multi-cam",
        "--splitting-algorithm", "transnetv2",
    ])
    import pytest as _pytest

    with _pytest.raises(ValueError, match="fixed-stride"):
        build_input_data(args2)


def test_driver_flag_aliases():
    """Reference flag spellings parse to the same dests."""
    import argparse

    from cosmos_curate_amd.pipelines.video.splitting_pipeline import _setup_parser

    p = argparse.ArgumentParser()
    _setup_parser(p)
    a = p.parse_args([
        "--input-video-path", "/i", "--output-clip-path", "/o",
        "--fixed-stride-min-clip-length-s", "5",
        "--no-generate-embeddings",
        "--motion-filter", "score-only",
    ])
    assert a.fixed_stride_min_clip_length == 5.0
    assert a.generate_embeddings is False
    assert a.motion_filter == "score-only"


def test_multicam_split_end_to_end_cpu(tmp_path):
    """Full multicam split on CPU (no embeddings): both cameras get the
    same clip spans/uuids and both cameras' clips are written."""
    import argparse
    import json
    import uuid as uuid_mod

    from cosmos_curate_amd.core.interfaces import SequentialRunner
    from cosmos_curate_amd.pipelines.video.splitting_pipeline import (
        _setup_parser,
        split,
    )
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    root = tmp_path / "sessions"
    sid = str(uuid_mod.uuid4())
    for cam in ["front", "rear"]:
        d = root / sid / cam
        d.mkdir(parents=True)
        (d / "v.nv12").write_bytes(
            raw_backend.make_synthetic_clip(600, 32, 48, 30, seed=3))
    out = tmp_path / "out"
    p = argparse.ArgumentParser()
    _setup_parser(p)
    args = p.parse_args([
        "--input-video-path", str(root), "--output-clip-path", str(out),
        "--multi-cam", "--no-embeddings",
    ])
    summary = split(args, runner=SequentialRunner())
    assert summary["num_input_videos"] == 2  # 2 cameras in 1 session
    assert summary["num_clips"] == 4  # 20 s x 2 cams -> 2 clips each
    assert summary["num_clips_with_errors"] == 0
    metas = sorted((out / "metas" / "v0").glob("*.json"))
    assert len(metas) == 4  # per-camera files (cam prefix), no collisions
    spans = {}
    for m in metas:
        meta = json.loads(m.read_text())
        spans.setdefault(meta["span"][0], []).append(meta["uuid"])
    # multicam cameras share the SAME clip uuid per span
    # (clip_extraction_stages.py:654-661)
    assert len(spans) == 2
    for uuids in spans.values():
        assert len(uuids) == 2 and uuids[0] == uuids[1]


def test_run_pipeline_unified_entry(tmp_path):
    """run_pipeline CLI-mode + config-mode dispatch (reference
    run_pipeline.py:17-27 semantics)."""
    import json as json_mod

    from cosmos_curate_amd.pipelines.video.run_pipeline import main
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    inp = tmp_path / "in"
    inp.mkdir()
    (inp / "v.nv12").write_bytes(raw_backend.make_synthetic_clip(300, 32, 48, 30, seed=2))

    # CLI mode
    s1 = main(["split", "--input-video-path", str(inp),
               "--output-clip-path", str(tmp_path / "o1"), "--no-embeddings"])
    assert s1["num_clips"] == 1

    # config mode (json)
    cfg = tmp_path / "job.json"
    cfg.write_text(json_mod.dumps({
        "pipeline": "split",
        "args": {"input_video_path": str(inp),
                 "output_clip_path": str(tmp_path / "o2"),
                 "no_embeddings": True},
    }))
    s2 = main([str(cfg)])
    assert s2["num_clips"] == 1

    # config mode (yaml)
    ycfg = tmp_path / "job.yaml"
    ycfg.write_text(
        "pipeline: split\n"
        "args:\n"
        f"  input_video_path: {inp}\n"
        f"  output_clip_path: {tmp_path / 'o3'}\n"
        "  no_embeddings: true\n"
    )
    s3 = main([str(ycfg)])
    assert s3["num_clips"] == 1

    import pytest as _pytest

    with _pytest.raises(SystemExit):
        main(["nope"])
