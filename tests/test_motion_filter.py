"""Motion filter math vs oracle + stage semantics (CPU)."""

import pathlib
import uuid

import numpy as np
import pytest

from cosmos_curate_amd.core.interfaces import SequentialRunner, run_pipeline
from cosmos_curate_amd.pipelines.video.filtering.motion.motion_filter_stages import (
    DecodedMotionData,
    MotionFilterStage,
    check_if_small_motion,
    motion_vectors_to_flowfield,
)
from cosmos_curate_amd.pipelines.video.utils.data_model import (
    Clip,
    SplitPipeTask,
    Video,
)
from oracle import motion as omotion


def make_mv(bw, bh, dst_x, dst_y, mx, my, scale=4.0):
    # [w, h, src_x, src_y, dst_x, dst_y, flags, motion_x, motion_y, scale]
    return [bw, bh, 0, 0, dst_x, dst_y, 0, mx, my, scale]


def test_flowfield_matches_oracle_no_overlap():
    import torch

    h, w = 64, 96
    mvs = np.array(
        [make_mv(16, 16, 16, 16, 8, -4), make_mv(8, 8, 52, 40, -12, 2),
         make_mv(16, 8, 80, 12, 3, 3), make_mv(8, 16, 30, 50, 0, -8)],
        dtype=np.float32,
    )
    want = omotion.motion_vectors_to_flowfield(mvs, (h, w))
    got = motion_vectors_to_flowfield(
        torch.from_numpy(mvs).unsqueeze(0), (h, w)
    )[0].numpy()
    np.testing.assert_allclose(got, want, atol=1e-6)
    # painted block has delta = -motion/scale
    np.testing.assert_allclose(want[16, 16], [-2.0, 1.0], atol=1e-6)


def test_scores_match_oracle():
    rng = np.random.default_rng(4)
    h, w = 256, 384
    frames = []
    for _ in range(12):
        n = rng.integers(5, 40)
        mvs = np.array(
            [
                make_mv(
                    *((8, 8) if rng.integers(2) else (16, 16)),
                    int(rng.integers(0, w)), int(rng.integers(0, h)),
                    int(rng.integers(-32, 33)), int(rng.integers(-32, 33)),
                )
                for _ in range(n)
            ],
            dtype=np.float32,
        )
        frames.append(mvs)
    small_o, patch_o, mean_o = omotion.check_if_small_motion(frames, (h, w))
    small_p, patch_p, mean_p = check_if_small_motion(frames, (h, w))
    # scatter overlap order is undefined upstream (module docstring):
    # score-level parity
    assert small_p == small_o
    assert mean_p == pytest.approx(mean_o, rel=2e-2)
    assert patch_p == pytest.approx(patch_o, rel=5e-2, abs=1e-7)


def test_motion_stage_filters_small_motion():
    def clip_with_motion(i, moving):
        c = Clip(uuid=uuid.uuid4(), source_video="s", span=(float(i), i + 1.0))
        if moving:
            frames = [
                np.array([make_mv(16, 16, x, 64, 40, 40) for x in range(8, 384, 16)],
                         dtype=np.float32)
                for _ in range(10)
            ]
        else:
            frames = [np.zeros((0, 10), dtype=np.float32) for _ in range(10)]
        c.decoded_motion_data = DecodedMotionData(frames, (128, 384))
        return c

    v = Video(input_video=pathlib.Path("/m.mp4"))
    v.clips = [clip_with_motion(0, True), clip_with_motion(1, False)]
    out = run_pipeline([SplitPipeTask(videos=[v])], [MotionFilterStage()],
                       runner=SequentialRunner())
    video = out[0].video
    assert len(video.clips) == 1 and len(video.filtered_clips) == 1
    kept = video.clips[0]
    assert kept.motion_score_global_mean > 0
    assert kept.decoded_motion_data is None  # payload freed
    filt = video.filtered_clips[0]
    assert filt.motion_score_global_mean == 0.0
    assert video.clip_stats.num_filtered_by_motion == 1  # per-filter counter


def test_missing_motion_data_records_error():
    v = Video(input_video=pathlib.Path("/m.mp4"))
    v.clips = [Clip(uuid=uuid.uuid4(), source_video="s", span=(0.0, 1.0))]
    out = run_pipeline([SplitPipeTask(videos=[v])], [MotionFilterStage()],
                       runner=SequentialRunner())
    clip = out[0].video.clips[0]
    assert clip.errors.get("motion") == "motion_decode_unavailable"
