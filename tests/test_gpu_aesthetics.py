"""Aesthetic filter on MI355X: scoring + filtering semantics."""

import pathlib
import uuid

import numpy as np
import pytest

from cosmos_curate_amd.core.interfaces import SequentialRunner, run_pipeline
from cosmos_curate_amd.core.utils.lazy_data import LazyData
from cosmos_curate_amd.pipelines.video.filtering.aesthetics.aesthetic_filter_stages import (
    AestheticFilterStage,
)
from cosmos_curate_amd.pipelines.video.utils.data_model import (
    Clip,
    SplitPipeTask,
    Video,
)

pytestmark = pytest.mark.gpu

SIG = "FrameExtractionPolicy.sequence-1000"


def make_task(n_clips=3):
    rng = np.random.default_rng(5)
    v = Video(input_video=pathlib.Path("/synthetic/a.mp4"))
    for i in range(n_clips):
        frames = rng.integers(0, 256, size=(4, 224, 224, 3), dtype=np.uint8)
        clip = Clip(
            uuid=uuid.uuid5(uuid.NAMESPACE_URL, f"aes{i}"),
            source_video="s",
            span=(float(i), float(i + 1)),
            encoded_data=np.ones(8, dtype=np.uint8),
        )
        clip.extracted_frames = LazyData(value={SIG: frames}, nbytes=frames.nbytes)
        v.clips.append(clip)
    return SplitPipeTask(videos=[v])


def test_scores_and_filtering():
    task = make_task()
    stage = AestheticFilterStage(score_threshold=1e9, target_fps=1.0)  # filter all
    out = run_pipeline([task], [stage], runner=SequentialRunner())
    video = out[0].video
    assert len(video.clips) == 0 and len(video.filtered_clips) == 3
    for clip in video.filtered_clips:
        assert clip.aesthetic_score is not None and clip.aesthetic_score < 1e9
        # last consumer dropped the frames dict
        assert clip.extracted_frames.resolve() is None

    task = make_task()
    stage = AestheticFilterStage(score_threshold=-1e9, target_fps=1.0)  # keep all
    out = run_pipeline([task], [stage], runner=SequentialRunner())
    video = out[0].video
    assert len(video.clips) == 3 and not video.filtered_clips
    assert video.clip_stats.num_passed == 3


def test_missing_signature_records_error():
    task = make_task(1)
    task.video.clips[0].extracted_frames = LazyData(value={"other": None}, nbytes=0)
    out = run_pipeline(
        [task], [AestheticFilterStage(score_threshold=0.0)], runner=SequentialRunner()
    )
    clip = out[0].video.filtered_clips[0]
    assert clip.aesthetic_score == -1.0
    assert any(k.startswith("frames-") for k in clip.errors)


def test_scorer_determinism():
    from cosmos_curate_amd.models.clip_aesthetics import CLIPAestheticScorer

    m = CLIPAestheticScorer()
    m.setup()
    rng = np.random.default_rng(9)
    frames = rng.integers(0, 256, size=(2, 224, 224, 3), dtype=np.uint8)
    s1 = m(frames).cpu().numpy()
    s2 = m(frames).cpu().numpy()
    np.testing.assert_array_equal(s1, s2)
    assert s1.shape == (2,)
