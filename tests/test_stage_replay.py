"""Stage save/replay/compare harness (SURVEY.md §4's A/B parity mechanism)."""

import pathlib

import numpy as np
import pytest

from cosmos_curate_amd.core.interfaces import SequentialRunner, run_pipeline
from cosmos_curate_amd.core.utils.stage_replay import (
    StageSaveConfig,
    compare_stage,
    load_saved,
    replay_stage,
)
from cosmos_curate_amd.pipelines.video.clipping.clip_extraction_stages import (
    FixedStrideExtractorStage,
)
from cosmos_curate_amd.pipelines.video.utils.data_model import (
    SplitPipeTask,
    Video,
    VideoMetadata,
)


def make_task():
    n = 720
    v = Video(
        input_video=pathlib.Path("/data/replay.mp4"),
        metadata=VideoMetadata(size=1, height=480, width=854, framerate=24.0,
                               num_frames=n, duration=30.0, video_codec="h264"),
        timestamps=(np.arange(n) / 24.0).astype(np.float32),
    )
    return SplitPipeTask(videos=[v])


def test_save_replay_compare_roundtrip(tmp_path):
    cfg = StageSaveConfig(output_path=str(tmp_path))
    out = run_pipeline(
        [make_task()], [FixedStrideExtractorStage()],
        runner=SequentialRunner(), stage_save_config=cfg,
    )
    assert len(out[0].video.clips) == 3

    saved_in = load_saved(str(tmp_path), "FixedStrideExtractorStage", "input")
    saved_out = load_saved(str(tmp_path), "FixedStrideExtractorStage", "output")
    assert len(saved_in) == 1 and len(saved_out) == 1
    assert len(saved_out[0][0].video.clips) == 3

    # replay reproduces outputs; compare passes at 100%
    res = compare_stage(FixedStrideExtractorStage(), str(tmp_path))
    assert res.pass_rate == 1.0 and res.total >= 6  # span + uuid per clip

    # a behavior change is caught
    with pytest.raises(AssertionError, match="stage-compare failed"):
        compare_stage(FixedStrideExtractorStage(clip_len_s=5.0, clip_stride_s=5.0,
                                                min_clip_length_s=5.0), str(tmp_path))


def test_selective_stage_wrap(tmp_path):
    cfg = StageSaveConfig(output_path=str(tmp_path), stages=["NotThisStage"])
    run_pipeline([make_task()], [FixedStrideExtractorStage()],
                 runner=SequentialRunner(), stage_save_config=cfg)
    assert not (tmp_path / "FixedStrideExtractorStage").exists()


def test_replay_stage_runs_setup(tmp_path):
    cfg = StageSaveConfig(output_path=str(tmp_path))
    run_pipeline([make_task()], [FixedStrideExtractorStage()],
                 runner=SequentialRunner(), stage_save_config=cfg)
    outs = replay_stage(FixedStrideExtractorStage(), str(tmp_path))
    assert len(outs) == 1 and len(outs[0][0].video.clips) == 3


def test_profiling_wrapper_cpu_backend(tmp_path):
    from cosmos_curate_amd.core.utils.profiling import (
        ProfilingConfig,
        profiling_wrapper,
    )

    stage = profiling_wrapper(
        FixedStrideExtractorStage(),
        ProfilingConfig(output_path=str(tmp_path), profile_cpu=True),
    )
    out = run_pipeline([make_task()], [stage], runner=SequentialRunner())
    assert len(out[0].video.clips) == 3
    dumps = list((tmp_path / "profile" / "FixedStrideExtractorStage").glob("cpu_*.pstats"))
    assert len(dumps) == 1
    import pstats

    st = pstats.Stats(str(dumps[0]))
    assert st.total_calls > 0


def test_profiling_wrapper_disabled_is_identity(tmp_path):
    from cosmos_curate_amd.core.utils.profiling import (
        ProfilingConfig,
        profiling_wrapper,
    )

    s = FixedStrideExtractorStage()
    assert profiling_wrapper(s, None) is s
    assert profiling_wrapper(s, ProfilingConfig(output_path=str(tmp_path))) is s
