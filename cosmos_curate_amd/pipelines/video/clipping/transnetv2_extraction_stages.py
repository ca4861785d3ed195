"""TransNetV2 clip extraction — the neural split path.

Mirror of /root/reference/cosmos_curate/pipelines/video/clipping/
transnetv2_extraction_stages.py (stage :39-213, windowing :215-261, scene
postprocess :264-412 — same constructor signature and defaults: threshold
0.4, min 2 s/48 frames, max 60 s stride mode, crop 0.5 s, 0.25 GPU) and of
``VideoFrameExtractionStage`` (frame_extraction_stages.py:71-204: whole
video decoded to (N,27,48,3) u8).

MI355X route: the tiny-res full-video decode reuses the fused NV12->RGB +
bilinear resize kernel at a 48x27 target; the
3D-conv net runs on torch-rocm (row 10).  Postprocess parity vs
oracle/transnet_post.py is exact; network parity vs the reference
implementation is pinned by tests/golden/transnetv2_golden.npz.
"""

from __future__ import annotations

import math
import uuid

import numpy as np
import numpy.typing as npt
import torch

from cosmos_curate_amd import hotpath
from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.lazy_data import LazyData
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.models import transnetv2
from cosmos_curate_amd.pipelines.video.utils import raw_backend
from cosmos_curate_amd.pipelines.video.utils.data_model import Clip, SplitPipeTask


# ---- windowing + scene postprocess (reference :215-412) -------------------

def _get_batches(frames: npt.NDArray[np.uint8]):
    """100-frame windows, 50-stride, 25-frame replicate padding (:215-235)."""
    total = len(frames)
    rem = -total % 50
    for i in range(0, total + rem, 50):
        lo, hi = max(i - 25, 0), min(i + 75, total)
        batch = frames[lo:hi]
        if i < 25:
            batch = np.concatenate([np.repeat(frames[:1], 25 - i, axis=0), batch], axis=0)
        if hi > total:
            batch = np.concatenate([batch, np.repeat(frames[-1:], hi - total, axis=0)], axis=0)
        yield batch


def _get_predictions(
    model, frames: npt.NDArray[np.uint8], threshold: float
) -> npt.NDArray[np.uint8]:
    """Batched forward; keep the central 50 frames of each window (:237-261)."""
    preds = []
    for batch in _get_batches(frames):
        t = torch.from_numpy(np.ascontiguousarray(batch))
        if torch.cuda.is_available():
            t = t.cuda()
        one_hot = model(t.unsqueeze(0))
        preds.append(one_hot[0, 25:75])
    stacked = torch.cat(preds, 0)[: len(frames)]
    return (stacked > threshold).to(torch.uint8).cpu().numpy()


def _get_scenes(
    predictions: npt.NDArray[np.uint8], *, entire_scene_as_clip: bool
) -> npt.NDArray[np.int32]:
    """0/1 flags -> scene [start,end) pairs (:264-296)."""
    scenes: list[tuple[int, int]] = []
    t_prev, start, t, i = 0, 0, -1, 0
    for i, t in enumerate(predictions):
        if t_prev == 1 and t == 0:
            start = i
        if t_prev == 0 and t == 1 and i != 0:
            scenes.append((start, i))
        t_prev = t
    if scenes and t == 0:
        scenes.append((start, i))
    if not scenes and entire_scene_as_clip:
        scenes.append((0, len(predictions)))
    return np.array(scenes, dtype=np.int32).reshape(-1, 2)


def _create_spans(start: int, end: int, max_length: int, min_length: int | None) -> list[list[int]]:
    """(:365-412)."""
    spans = []
    pos = start
    while pos < end:
        stop = min(pos + max_length, end)
        if min_length and (stop - pos) < min_length and stop == end:
            break
        spans.append([pos, stop])
        pos = stop
    return spans


def _crop_scenes(scenes: npt.NDArray[np.int32], crop_length: int) -> npt.NDArray[np.int32]:
    """(:348-363)."""
    cropped = np.stack([scenes[:, 0] + crop_length, scenes[:, 1] - crop_length]).T
    return cropped[(cropped[:, 1] - cropped[:, 0]) > 0]


def _get_filtered_scenes(
    scenes: npt.NDArray[np.int32],
    min_length: int | None = None,
    max_length: int | None = None,
    max_length_mode: str = "truncate",
    crop_length: int | None = None,
) -> npt.NDArray[np.int32]:
    """(:296-346)."""
    scenes = scenes.copy()
    if max_length is not None:
        if max_length_mode == "truncate":
            scenes[:, 1] = np.minimum(scenes[:, 0] + max_length, scenes[:, 1])
        elif max_length_mode == "stride":
            out: list[list[int]] = []
            for s, e in scenes:
                out.extend(_create_spans(int(s), int(e), max_length, min_length))
            scenes = np.array(out, dtype=scenes.dtype).reshape(-1, 2)
        else:
            raise NotImplementedError(max_length_mode)
    if crop_length is not None:
        scenes = _crop_scenes(scenes, crop_length)
    if min_length is not None:
        scenes = scenes[(scenes[:, 1] - scenes[:, 0]) >= min_length]
    return scenes


# ---- whole-video tiny-res frame extraction (frame_extraction_stages.py) ---

class VideoFrameExtractionStage(CuratorStage):
    """Whole-video decode to (N,27,48,3) u8 (frame_extraction_stages.py:71).

    Raw-NV12 payloads run the fused HIP NV12->RGB+resize kernel at the
    27x48 target; H.264 needs rocDecode (recorded error when absent).
    """

    def __init__(self, *, num_cpus_per_worker: float = 3.0,
                 verbose: bool = False, log_stats: bool = False) -> None:
        self._timer = StageTimer(self)
        self._num_cpus = num_cpus_per_worker
        self._verbose = verbose
        self._log_stats = log_stats

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=self._num_cpus, gpus=0.25)

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        lib = hotpath.require_gpu()
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            for video in task.videos:
                with self._timer.time_process():
                    data = video.encoded_data.resolve()
                    if data is None:
                        video.errors["frame_extraction"] = "no encoded data"
                        continue
                    raw = data  # raw_backend consumes any buffer (no copy)
                    if not raw_backend.is_raw_nv12(raw):
                        video.errors["frame_extraction"] = "decode_unavailable"
                        continue
                    n, h, w, _fps = raw_backend.parse_header(raw)
                    idx = np.arange(n, dtype=np.int32)
                    ys, uvs = raw_backend.frame_planes(raw, idx)
                    dev = torch.device("cuda")
                    y_dev = torch.from_numpy(ys).to(dev)
                    uv_dev = torch.from_numpy(uvs).to(dev)
                    out = torch.empty((n, 27, 48, 3), dtype=torch.uint8, device=dev)
                    stream = torch.cuda.current_stream(dev).cuda_stream
                    hotpath.check(
                        lib.cc_nv12_to_rgb_resize(
                            y_dev.data_ptr(), uv_dev.data_ptr(), n, h, w, w,
                            out.data_ptr(), 27, 48, stream,
                        )
                    )
                    arr = out.cpu().numpy()
                    video.frame_array = LazyData(value=arr, nbytes=arr.nbytes)
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks


class TransNetV2ClipExtractionStage(CuratorStage):
    """Neural shot-boundary clip extraction (reference :39-213)."""

    def __init__(
        self,
        threshold: float = 0.4,
        min_length_s: float | None = 2.0,
        min_length_frames: int | None = 48,
        max_length_s: float | None = 60.0,
        max_length_mode: str = "stride",
        crop_s: float | None = 0.5,
        *,
        entire_scene_as_clip: bool = True,
        num_gpus_per_worker: float = 0.25,
        limit_clips: int = 0,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self.threshold = threshold
        self.min_length_s = min_length_s
        self.min_length_frames = min_length_frames
        self.max_length_s = max_length_s
        if self.min_length_s and self.max_length_s and self.max_length_s < self.min_length_s:
            msg = "Max length is smaller than min length!"
            raise ValueError(msg)
        self.max_length_mode = max_length_mode
        self.crop_s = crop_s
        self.entire_scene_as_clip = entire_scene_as_clip
        self._num_gpus_per_worker = num_gpus_per_worker
        self._limit_clips = limit_clips
        self._verbose = verbose
        self._log_stats = log_stats
        self._model = transnetv2.TransNetV2()

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(gpus=self._num_gpus_per_worker)

    @property
    def model(self) -> ModelInterface:
        return self._model

    def _get_min_length(self, framerate: float) -> int | None:
        ml = math.ceil(self.min_length_s * framerate) if self.min_length_s is not None else None
        if self.min_length_frames is not None:
            ml = max(ml, self.min_length_frames) if ml is not None else self.min_length_frames
        return ml

    def _get_max_length(self, framerate: float) -> int | None:
        return math.ceil(self.max_length_s * framerate) if self.max_length_s is not None else None

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            video = task.video
            src = video.input_video
            if not video.has_metadata() or not video.frame_array:
                continue
            assert video.metadata.framerate
            with self._timer.time_process():
                frames = video.frame_array.resolve()
                if tuple(frames.shape[1:4]) != (27, 48, 3):
                    msg = f"Expected frames of shape 27x48x3, got {frames.shape[1:4]}."
                    raise ValueError(msg)
                predictions = _get_predictions(self._model, frames, self.threshold)
                scenes = _get_scenes(predictions, entire_scene_as_clip=self.entire_scene_as_clip)
                filtered = _get_filtered_scenes(
                    scenes,
                    min_length=self._get_min_length(video.metadata.framerate),
                    max_length=self._get_max_length(video.metadata.framerate),
                    max_length_mode=self.max_length_mode,
                    crop_length=(int(self.crop_s * video.metadata.framerate) if self.crop_s else None),
                )
                for start_f, end_f in filtered:
                    # uuid5 over "{src}_{start_frame}_{end_frame}" (ref :193)
                    clip = Clip(
                        uuid=uuid.uuid5(uuid.NAMESPACE_URL, f"{src}_{start_f}_{end_f}"),
                        source_video=str(src),
                        span=(
                            float(start_f) / video.metadata.framerate,
                            float(end_f) / video.metadata.framerate,
                        ),
                    )
                    video.clips.append(clip)
                    if self._limit_clips > 0 and len(video.clips) >= self._limit_clips:
                        break
                video.frame_array.drop()
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
