"""CLIP embedding stages — the rebuild's embedder pair.

Mirror of the reference's two-stage embedding pattern
(/root/reference/cosmos_curate/pipelines/video/embedding/
internvideo2_stages.py: FrameCreation :43-185 + Embedding :187-300), with
CLIP-ViT-B/32 as the model (the embedder BASELINE.json names for configs
#1/#2; IV2/CE1 weights are unavailable offline — SURVEY.md §2 row
"Embedding models").

- ``ClipFrameCreationStage``: picks the extracted-frame array for the
  model's signature (internvideo2_stages.py:117-135 lookup semantics) and
  stores it for the embedder; on the GPU route frames are already device
  tensors, so "frame creation" is a key lookup + optional temporal
  subsample, NOT a numpy resize pass (that work fused into the
  extraction kernel).
- ``ClipEmbeddingStage``: batched ViT forward (batch 8 clips, 0.25 GPU —
  embedding_builders.py:67-76 defaults), frames preprocessed by the fused
  HIP normalize kernel, per-frame embeddings mean-pooled and
  L2-normalized into ``clip.clip_embedding`` (the rebuild's counterpart
  of intern_video_2_embedding).
"""

from __future__ import annotations

import numpy as np
import torch

from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.core.utils.lazy_data import LazyData
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.core.utils.roctx import annotate
from cosmos_curate_amd.models.clip import CLIPImageEmbeddings
from cosmos_curate_amd.pipelines.video.clipping.clip_frame_extraction_stages import (
    extract_frames,
)
from cosmos_curate_amd.pipelines.video.utils.data_model import Clip, SplitPipeTask
from cosmos_curate_amd.pipelines.video.utils.decoder_utils import (
    FrameExtractionPolicy,
    FrameExtractionSignature,
)


class ClipFrameCreationStage(CuratorStage):
    """Select frames for the embedder (internvideo2_stages.py:43-185 shape)."""

    def __init__(
        self,
        target_fps: float = 2.0,
        *,
        max_frames: int | None = None,
        min_frames: int | None = None,
        target_res: tuple[int, int] | None = None,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._target_fps = target_fps
        self._max_frames = max_frames
        self._min_frames = min_frames
        # re-extraction must regenerate frames at the SAME resolution the
        # extraction stage produced, or the embedder sees mixed sizes
        self._target_res = target_res if target_res is not None else (-1, -1)
        self._verbose = verbose
        self._log_stats = log_stats

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=1.0)

    def _process_clip(self, clip: Clip) -> None:
        sig = FrameExtractionSignature(
            FrameExtractionPolicy.sequence, self._target_fps
        ).to_str()
        frames_map = clip.extracted_frames.resolve()
        if not frames_map or sig not in frames_map:
            # signature mismatch = silent missing-frames path upstream;
            # here it is a recorded per-clip error
            clip.errors["clip_embedding_frames"] = f"missing signature {sig}"
            return
        frames = frames_map[sig]
        # frame-count guarantee: re-extract at doubling fps (<=20) until the
        # model's required count is reached (internvideo2_stages.py:137-175;
        # on running out of fps headroom the reference logs and keeps what it
        # has — no per-clip error)
        if self._min_frames is not None and len(frames) < self._min_frames:
            data = clip.encoded_data.resolve()
            regen_fps = self._target_fps
            while data is not None and len(frames) < self._min_frames:
                regen_fps *= 2
                if regen_fps > 20:
                    break
                raw = data  # buffers pass through; no bytes() copy
                frames = extract_frames(
                    raw, sample_rate_fps=regen_fps,
                    target_res=self._target_res,
                    to_host=isinstance(frames, np.ndarray),
                )
        if self._max_frames is not None and len(frames) > self._max_frames:
            step = len(frames) // self._max_frames  # IV2 step rule (:400-401)
            frames = frames[::step][: self._max_frames]
        nbytes = frames.nbytes if isinstance(frames, np.ndarray) else frames.numel()
        clip.clip_embedding_frames = LazyData(value=frames, nbytes=nbytes)

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                for video in task.videos:
                    for clip in video.clips:
                        self._process_clip(clip)
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks


class ClipEmbeddingStage(CuratorStage):
    """Batched CLIP ViT embed (internvideo2_stages.py:187-300 shape)."""

    def __init__(
        self,
        num_gpus_per_worker: float = 0.25,
        *,
        batch_size: int = 8,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._num_gpus = num_gpus_per_worker
        self._batch_size = batch_size
        self._verbose = verbose
        self._log_stats = log_stats
        self._model = CLIPImageEmbeddings()

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(gpus=self._num_gpus)

    @property
    def model(self) -> ModelInterface:
        return self._model

    def _embed_clips(self, clips: list[Clip]) -> None:
        """One batched forward over the frames of up to batch_size clips."""
        payloads: list[tuple[Clip, object]] = []
        for clip in clips:
            frames = clip.clip_embedding_frames.resolve()
            if frames is None or len(frames) == 0:
                clip.errors["clip_embedding"] = "no frames"
                continue
            payloads.append((clip, frames))
        if not payloads:
            return
        counts = [len(f) for _, f in payloads]
        if isinstance(payloads[0][1], np.ndarray):
            batch = np.concatenate([f for _, f in payloads], axis=0)
        else:
            batch = torch.cat([f for _, f in payloads], dim=0)
        embeds = self._model(batch)  # (sum(counts), 512) f32, unit-norm
        pos = 0
        for (clip, _), cnt in zip(payloads, counts):
            e = embeds[pos : pos + cnt].mean(dim=0)
            e = e / torch.linalg.vector_norm(e)
            clip.clip_embedding = e.cpu().numpy().astype(np.float32)
            pos += cnt

    @annotate("ClipEmbeddingStage")
    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                for video in task.videos:
                    todo = [c for c in video.clips if not c.errors]
                    for i in range(0, len(todo), self._batch_size):
                        try:
                            self._embed_clips(todo[i : i + self._batch_size])
                        except Exception as e:
                            for c in todo[i : i + self._batch_size]:
                                c.errors["clip_embedding"] = str(e)
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
