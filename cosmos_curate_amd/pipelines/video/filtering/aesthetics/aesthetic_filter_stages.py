"""Aesthetic score filter stage.

Mirror of /root/reference/cosmos_curate/pipelines/video/filtering/
aesthetics/aesthetic_filter_stages.py:41-220 (``AestheticFilterStage``):
same constructor signature (score_threshold, target_fps=1.0,
reduction="min"), same frame-signature pop semantics (each consumer pops
its key; last consumer drops the LazyData), same filtered_clips routing
and ClipStats accounting.  Scoring runs on the shared MFMA CLIP tower +
seeded MLP head (models/clip_aesthetics.py).
"""

from __future__ import annotations

import numpy as np

from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.models.clip_aesthetics import CLIPAestheticScorer
from cosmos_curate_amd.pipelines.video.utils.data_model import SplitPipeTask
from cosmos_curate_amd.pipelines.video.utils.decoder_utils import (
    FrameExtractionPolicy,
    FrameExtractionSignature,
)


class AestheticFilterStage(CuratorStage):
    """aesthetic_filter_stages.py:41: score clips, filter below threshold."""

    def __init__(
        self,
        score_threshold: float = 3.5,
        target_fps: float = 1.0,
        reduction: str = "min",
        *,
        num_gpus_per_worker: float = 0.25,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._score_threshold = score_threshold
        self._reduction = reduction
        self._num_gpus_per_worker = num_gpus_per_worker
        self._verbose = verbose
        self._log_stats = log_stats
        self._frame_extraction_signature = FrameExtractionSignature(
            FrameExtractionPolicy.sequence, target_fps
        ).to_str()
        self._model = CLIPAestheticScorer()
        self._reduce_fn = None

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(gpus=self._num_gpus_per_worker)

    @property
    def model(self) -> ModelInterface:
        return self._model

    def stage_setup(self) -> None:
        self._model.setup()
        if self._reduction == "mean":
            self._reduce_fn = np.mean
        elif self._reduction == "min":
            self._reduce_fn = np.min
        else:
            msg = f"Reduction `{self._reduction}` not implemented."
            raise NotImplementedError(msg)

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                for video in task.videos:
                    passed = []
                    for clip in video.clips:
                        if not clip.encoded_data:
                            clip.errors["encoded_data"] = "empty"
                            clip.aesthetic_score = -1.0
                        else:
                            ef = clip.extracted_frames.resolve()
                            if ef is None or self._frame_extraction_signature not in ef:
                                clip.errors[
                                    f"frames-{self._frame_extraction_signature}"
                                ] = "missing"
                                clip.aesthetic_score = -1.0
                            else:
                                frames = ef.pop(self._frame_extraction_signature)
                                scores = self._model(frames).cpu().numpy()
                                clip.aesthetic_score = float(self._reduce_fn(scores))
                                if not ef:
                                    clip.extracted_frames.drop()  # last consumer
                        if (
                            clip.aesthetic_score is not None
                            and clip.aesthetic_score >= self._score_threshold
                        ):
                            passed.append(clip)
                            video.clip_stats.num_passed += 1
                        else:
                            video.filtered_clips.append(clip)
                            video.clip_stats.num_filtered += 1
                            video.clip_stats.num_filtered_by_aesthetic += 1
                    video.clips = passed
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
