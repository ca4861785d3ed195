"""Motion-vector filter arithmetic (CPU oracle).

Restates /root/reference/cosmos_curate/pipelines/video/filtering/motion/
motion_vector_backend.py:
- ``motion_vectors_to_flowfield`` (:85-168): paint each MV's block with
  delta = -motion/motion_scale at its dst-centred block (sizes 8x8/16x16/
  16x8/8x16, clamped), batch of flow images.  NOTE: the reference writes
  overlapping blocks through torch.index_put_(accumulate=False) whose
  duplicate-index order is UNDEFINED — overlap resolution is therefore not
  a bit-contract upstream; this restatement paints sequentially in MV
  order (later MV wins) and parity with the product is pinned at the
  score level (rtol), not per-pixel.
- ``check_if_small_motion`` (:253-313): magnitudes |flow|/(H+W), global
  mean over frames*H*W, per-pixel temporal mean -> 1/256 bilinear
  downscale (cv2.resize INTER_LINEAR semantics) -> min; thresholds
  global 0.00098 / per-patch 1e-6.

MV rows use the ffmpeg AVMotionVector field order AFTER the reference's
`[:, 1:]` slice (:295): [w, h, src_x, src_y, dst_x, dst_y, flags,
motion_x, motion_y, motion_scale].
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt

BLOCK_OPTIONS = ((8, 8), (16, 16), (16, 8), (8, 16))


def motion_vectors_to_flowfield(
    mvs: npt.NDArray[np.float32], size: tuple[int, int]
) -> npt.NDArray[np.float32]:
    """(n_vectors, 10) f32 -> (H, W, 2) flow for ONE frame."""
    h, w = size
    flow = np.zeros((h, w, 2), dtype=np.float32)
    for row in mvs:
        bw, bh = int(row[0]), int(row[1])
        if (bw, bh) not in BLOCK_OPTIONS:
            continue
        dst_x, dst_y = int(row[4]), int(row[5])
        scale = row[9] if row[9] != 0 else 1.0
        delta = np.array([-row[7] / scale, -row[8] / scale], dtype=np.float32)
        # block extent centred on dst: offsets [-b//2, b//2) per axis, with
        # the reference's clamp-to-edge on both ends (:146-151)
        xs = np.clip(dst_x + np.arange(-(bw // 2), bw // 2), 0, w - 1)
        ys = np.clip(dst_y + np.arange(-(bh // 2), bh // 2), 0, h - 1)
        for y in ys:
            flow[y, xs] = delta
    return flow


def _bilinear_resize_f32(img: npt.NDArray[np.float32], out_h: int, out_w: int):
    """cv2 INTER_LINEAR pixel-center resize on a float image."""
    h, w = img.shape
    sy = (np.arange(out_h) + 0.5) * (h / out_h) - 0.5
    sx = (np.arange(out_w) + 0.5) * (w / out_w) - 0.5
    y0 = np.clip(np.floor(sy), 0, h - 1).astype(np.int64)
    x0 = np.clip(np.floor(sx), 0, w - 1).astype(np.int64)
    y1 = np.minimum(y0 + 1, h - 1)
    x1 = np.minimum(x0 + 1, w - 1)
    wy = np.clip(sy - y0, 0, 1).astype(np.float32)
    wx = np.clip(sx - x0, 0, 1).astype(np.float32)
    top = img[y0][:, x0] * (1 - wx)[None, :] + img[y0][:, x1] * wx[None, :]
    bot = img[y1][:, x0] * (1 - wx)[None, :] + img[y1][:, x1] * wx[None, :]
    return top * (1 - wy)[:, None] + bot * wy[:, None]


def check_if_small_motion(
    mv_list: list[npt.NDArray[np.float32]],
    frame_shape: tuple[int, int],
    global_mean_threshold: float = 0.00098,
    per_patch_min_256_threshold: float = 0.000001,
) -> tuple[bool, float, float]:
    """Returns (is_small_motion, per_patch_min_256, global_mean)."""
    h, w = frame_shape
    per_pixel_sum = np.zeros((h, w), dtype=np.float64)
    global_sum = 0.0
    for mv in mv_list:
        flow = motion_vectors_to_flowfield(mv.astype(np.float32), (h, w))
        mag = np.linalg.norm(flow, axis=2) / (h + w)
        global_sum += float(mag.sum())
        per_pixel_sum += mag
    n = max(len(mv_list), 1)
    global_mean = global_sum / (n * h * w)
    per_pixel_avg = (per_pixel_sum / n).astype(np.float32)
    oh, ow = max(1, round(h / 256)), max(1, round(w / 256))
    per_patch_min = float(_bilinear_resize_f32(per_pixel_avg, oh, ow).min())
    small = global_mean < global_mean_threshold or per_patch_min < per_patch_min_256_threshold
    return small, per_patch_min, global_mean
