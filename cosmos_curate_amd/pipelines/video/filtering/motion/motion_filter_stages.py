"""Motion filter — small-motion clip rejection.

Mirror of /root/reference/cosmos_curate/pipelines/video/filtering/motion/
(motion_vector_backend.py flowfield/score math :85-313; stages :40-220):
``MotionFilterStage`` scores clips by painted motion-vector flow fields
(global mean + per-patch-256 minimum of |flow|/(H+W)) and moves
small-motion clips to ``filtered_clips``.

MI355X status of the two halves:
- the SCORE math runs here on torch (GPU when available) mirroring the
  reference's tensor pipeline; its scatter paint
  (torch.index_put_(accumulate=False)) has UNDEFINED duplicate-index order
  upstream, so parity with oracle/motion.py is pinned at score level.  The
  paint is bandwidth-trivial (<< 1 % of a clip's budget) — no dedicated
  HIP kernel until the semantics are worth pinning tighter.
- MV EXTRACTION needs decoder side-data (decode_for_motion,
  motion_vector_backend.py:169-250 — the export_mvs decoder flag): that is
  the same rocDecode seam as frame decode.  Clips carry
  pre-extracted MV arrays in ``decoded_motion_data``; mp4 clips without a
  decoder record ``motion_decode_unavailable``.
"""

from __future__ import annotations

import dataclasses

import numpy as np
import numpy.typing as npt
import torch

from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.pipelines.video.utils.data_model import SplitPipeTask

BLOCK_OPTIONS = ((8, 8), (16, 16), (16, 8), (8, 16))


@dataclasses.dataclass
class DecodedMotionData:
    """motion_vector_backend.py:55-81 shape: per-frame MV arrays + size."""

    frames: list[npt.NDArray[np.float32]]
    frame_size: tuple[int, int]

    def get_major_size(self) -> int:
        return sum(f.nbytes for f in self.frames)


def motion_vectors_to_flowfield(
    mvs: torch.Tensor, size: tuple[int, int], flow: torch.Tensor | None = None
) -> torch.Tensor:
    """(B, n, 10) -> (B, H, W, 2), the reference's paint (:85-168)."""
    h, w = size
    b = mvs.shape[0]
    device = mvs.device
    if flow is None or flow.shape != (b, h, w, 2):
        flow = torch.zeros(b, h, w, 2, device=device)
    else:
        flow.zero_()
    block_sizes = mvs[..., 0:2]
    dst = mvs[..., 4:6]
    motion = mvs[..., 7:9]
    scale = mvs[..., 9].unsqueeze(-1)
    delta = -motion / torch.where(scale == 0, torch.ones_like(scale), scale)
    for bw, bh in BLOCK_OPTIONS:
        sel = (block_sizes == torch.tensor([bw, bh], device=device)).all(-1)
        if not sel.any():
            continue
        bidx, vidx = sel.nonzero(as_tuple=True)
        d = dst[bidx, vidx]  # (k, 2) x,y
        dl = delta[bidx, vidx]  # (k, 2)
        offx = torch.arange(-(bw // 2), bw // 2, device=device)
        offy = torch.arange(-(bh // 2), bh // 2, device=device)
        xs = (d[:, 0:1] + offx).clamp_(0, w - 1).long()  # (k, bw)
        ys = (d[:, 1:2] + offy).clamp_(0, h - 1).long()  # (k, bh)
        bb = bidx.view(-1, 1, 1).expand(-1, bh, bw)
        yy = ys.view(-1, bh, 1).expand(-1, -1, bw)
        xx = xs.view(-1, 1, bw).expand(-1, bh, -1)
        vals = dl.view(-1, 1, 1, 2).expand(-1, bh, bw, 2)
        flow.index_put_((bb.reshape(-1), yy.reshape(-1), xx.reshape(-1)),
                        vals.reshape(-1, 2), accumulate=False)
    return flow


def check_if_small_motion(
    mv_list: list[npt.NDArray[np.float32]],
    frame_shape: tuple[int, int],
    global_mean_threshold: float = 0.00098,
    per_patch_min_256_threshold: float = 0.000001,
    *,
    use_gpu: bool = False,
    batch_size: int = 256,
) -> tuple[bool, float, float]:
    """motion_vector_backend.py:253-313 semantics."""
    h, w = frame_shape
    device = torch.device("cuda" if use_gpu else "cpu")
    global_sum = torch.tensor(0.0, device=device, dtype=torch.float64)
    per_pixel = torch.zeros((h, w), device=device, dtype=torch.float64)
    n = 0
    for off in range(0, len(mv_list), batch_size):
        chunk = mv_list[off : off + batch_size]
        maxv = max(m.shape[0] for m in chunk)
        padded = torch.zeros(len(chunk), max(maxv, 1), 10, dtype=torch.float32, device=device)
        for i, m in enumerate(chunk):
            if len(m):
                padded[i, : m.shape[0]] = torch.as_tensor(m, dtype=torch.float32, device=device)
        flow = motion_vectors_to_flowfield(padded, (h, w))
        mag = torch.linalg.vector_norm(flow, dim=3) / (h + w)
        global_sum += mag.sum().double()
        per_pixel += mag.sum(dim=0).double()
        n += len(chunk)
    global_mean = float(global_sum.item() / max(n * h * w, 1))
    per_pixel_avg = (per_pixel / max(n, 1)).float()
    oh, ow = max(1, round(h / 256)), max(1, round(w / 256))
    patches = torch.nn.functional.interpolate(
        per_pixel_avg.view(1, 1, h, w), size=(oh, ow), mode="bilinear",
        align_corners=False,
    )
    per_patch_min = float(patches.min().item())
    small = (global_mean < global_mean_threshold
             or per_patch_min < per_patch_min_256_threshold)
    return small, per_patch_min, global_mean


class MotionFilterStage(CuratorStage):
    """motion_filter_stages.py:40-220 shape: score + filter clips."""

    def __init__(
        self,
        global_mean_threshold: float = 0.00098,
        per_patch_min_256_threshold: float = 0.000001,
        *,
        score_only: bool = False,
        num_gpus_per_worker: float = 0.25,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._global_mean_threshold = global_mean_threshold
        self._per_patch_threshold = per_patch_min_256_threshold
        self._score_only = score_only
        self._num_gpus = num_gpus_per_worker
        self._verbose = verbose
        self._log_stats = log_stats

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(gpus=self._num_gpus)

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        use_gpu = torch.cuda.is_available()
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                for video in task.videos:
                    kept = []
                    for clip in video.clips:
                        md = clip.decoded_motion_data
                        if md is None or not md.frames:
                            clip.errors["motion"] = "motion_decode_unavailable"
                            kept.append(clip)
                            continue
                        small, patch_min, gmean = check_if_small_motion(
                            md.frames, md.frame_size,
                            self._global_mean_threshold,
                            self._per_patch_threshold,
                            use_gpu=use_gpu,
                        )
                        clip.motion_score_global_mean = gmean
                        clip.motion_score_per_patch_min_256 = patch_min
                        clip.decoded_motion_data = None  # free MV payload
                        if small and not self._score_only:
                            video.filtered_clips.append(clip)
                            video.clip_stats.num_filtered += 1
                            video.clip_stats.num_filtered_by_motion += 1
                        else:
                            kept.append(clip)
                    video.clips = kept
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
