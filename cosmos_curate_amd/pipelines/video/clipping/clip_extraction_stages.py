"""Fixed-stride clip extraction.

Mirror of /root/reference/cosmos_curate/pipelines/video/clipping/
clip_extraction_stages.py, hot-path subset:
- ``_make_spans_fixed_stride``      (:512-551)
- ``_make_clip_uuids``              (:554-565)  uuid5 contract
- ``_populate_clips_fixed_stride``  (:568-661)
- ``FixedStrideExtractorStage``     (:664-744)  same constructor signature,
  defaults (10 s len/stride/min, limit 0) and resources (1 CPU).

Bit-exact span + UUID parity vs oracle/spans.py is pinned in
tests/test_product_vs_oracle.py; golden expectations in
tests/golden/spans_kats.json.
"""

from __future__ import annotations

import uuid

from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.pipelines.video.utils.data_model import (
    Clip,
    SplitPipeTask,
    Video,
)


def _make_spans_fixed_stride(
    start_s: float,
    end_s: float,
    clip_len_s: float,
    clip_stride_s: float,
    min_clip_length_s: float,
) -> list[tuple[float, float]]:
    """Fixed-stride spans (clip_extraction_stages.py:512-551)."""
    spans: list[tuple[float, float]] = []
    pos = start_s
    while pos < end_s:
        end = min(pos + clip_len_s, end_s)
        if (end - pos) >= min_clip_length_s:
            spans.append((pos, end))
        pos += clip_stride_s
    return spans


def _make_clip_uuids(session_id: str, spans: list[tuple[float, float]]) -> list[uuid.UUID]:
    """uuid5(NAMESPACE_URL, f"{session}_{s}_{e}") (clip_extraction_stages.py:554-565)."""
    return [uuid.uuid5(uuid.NAMESPACE_URL, f"{session_id}_{s}_{e}") for (s, e) in spans]


def _get_videos_durations(videos: list[Video]) -> list[float]:
    """num_frames/framerate duration convention (clip_extraction_stages.py:486-509)."""

    def one(v: Video) -> float:
        nf, fr = v.metadata.num_frames, v.metadata.framerate
        if nf is None or fr is None or fr <= 0:
            return -1.0
        return float(nf / fr)

    return [one(v) for v in videos]


def _populate_clips_fixed_stride(
    videos: list[Video],
    session_id: str,
    clip_len_s: float,
    clip_stride_s: float,
    min_clip_length_s: float,
    *,
    limit_clips: int = 0,
) -> None:
    """Populate video.clips in place (clip_extraction_stages.py:568-661).

    Preserves the reference's start=0 convention: spans run over
    [0, min(durations)) regardless of first-PTS offset (:633-635).
    """
    durations = _get_videos_durations(videos)
    if any(d <= 0 for d in durations):
        msg = "Some videos have invalid (zero or negative) duration"
        raise ValueError(msg)
    for v in videos:
        if v.timestamps is None or len(v.timestamps) == 0:
            msg = f"Video {v.input_path} has no timestamps"
            raise ValueError(msg)
    end_s = min(durations)
    spans = _make_spans_fixed_stride(0.0, end_s, clip_len_s, clip_stride_s, min_clip_length_s)
    if limit_clips > 0:
        spans = spans[:limit_clips]
    clip_uuids = _make_clip_uuids(session_id, spans)
    for span, cu in zip(spans, clip_uuids):
        for video in videos:
            video.clips.append(
                Clip(uuid=cu, source_video=str(video.input_video), span=span)
            )


class FixedStrideExtractorStage(CuratorStage):
    """clip_extraction_stages.py:664-744: split videos into fixed clips."""

    def __init__(
        self,
        clip_len_s: float = 10,
        clip_stride_s: float = 10,
        min_clip_length_s: float = 10,
        limit_clips: int = 0,
        *,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self.clip_len_s = clip_len_s
        self.clip_stride_s = clip_stride_s
        self.min_clip_length_s = min_clip_length_s
        self._limit_clips = limit_clips
        self._verbose = verbose
        self._log_stats = log_stats

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=1.0)

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                try:
                    session_id = task.video.input_path
                    _populate_clips_fixed_stride(
                        task.videos,
                        session_id,
                        self.clip_len_s,
                        self.clip_stride_s,
                        self.min_clip_length_s,
                        limit_clips=self._limit_clips,
                    )
                except Exception as e:  # per-item error convention (§8b)
                    for video in task.videos:
                        video.errors[type(self).__name__] = str(e)
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
