"""InternVideo2 / Cosmos-Embed1 input-frame preparation (MI355X route).

The IV2/CE1 model WEIGHTS are unavailable offline, so their end-model
numerics stay parity-unpinned this round (SURVEY.md §2 row "Embedding
models" / §8c); what the rebuild ships — per that row's contract — is the
exact PREPROCESSING math their stages feed the networks with:

- ``formulate_input_frames`` mirrors internvideo2_mm.py:391-438
  ``_construct_frames``: temporal subsample step = len//fnum then [:fnum],
  per-frame bilinear resize to (size, size) (cv2.resize default
  INTER_LINEAR, :402), (x/255 - mean)/std with the ImageNet constants
  (:378-379), stacked to (1, fnum, 3, H, W) f32.
- on GPU the resize runs the bilinear HIP kernel and the normalize runs
  the fused preprocess kernel with the IV2 constants (both kernels take
  mean/std as arguments); on host arrays the same arithmetic is applied
  through the stage's device path — there is no CPU product fallback.

Parity: oracle/iv2_preprocess.py restates the same math in numpy;
tests pin product-GPU == oracle bit-exact on u8->resize and to f32
round-off on normalization.
"""

from __future__ import annotations

import ctypes

import numpy as np
import numpy.typing as npt
import torch

from cosmos_curate_amd import hotpath

IV2_MEAN = (0.485, 0.456, 0.406)  # internvideo2_mm.py:378
IV2_STD = (0.229, 0.224, 0.225)   # internvideo2_mm.py:379


def temporal_subsample(n_frames: int, fnum: int) -> npt.NDArray[np.int64]:
    """Frame indices of the IV2 rule: [::len//fnum][:fnum] (:400-401)."""
    if n_frames < fnum:
        msg = f"Frame count {n_frames} is smaller than minimal requirement {fnum}"
        raise ValueError(msg)
    step = n_frames // fnum
    return np.arange(n_frames)[::step][:fnum]


@torch.no_grad()
def formulate_input_frames(
    frames_u8_dev: torch.Tensor, fnum: int = 8, target_size: int = 224
) -> torch.Tensor:
    """(T,H,W,3) u8 CUDA tensor -> (1, fnum, 3, size, size) f32 CUDA.

    The device mirror of internvideo2_mm.formulate_input_frames (:426-438).
    """
    lib = hotpath.require_gpu()
    assert frames_u8_dev.is_cuda and frames_u8_dev.dtype == torch.uint8
    t, h, w, _ = frames_u8_dev.shape
    sel = temporal_subsample(t, fnum)
    picked = frames_u8_dev[torch.as_tensor(sel, device=frames_u8_dev.device)].contiguous()
    stream = torch.cuda.current_stream(frames_u8_dev.device).cuda_stream
    if (h, w) != (target_size, target_size):
        resized = torch.empty(
            (fnum, target_size, target_size, 3), dtype=torch.uint8,
            device=frames_u8_dev.device,
        )
        hotpath.check(
            lib.cc_resize_bilinear_u8(
                picked.data_ptr(), fnum, h, w, resized.data_ptr(),
                target_size, target_size, stream,
            )
        )
    else:
        resized = picked
    out = torch.empty(
        (fnum, 3, target_size, target_size), dtype=torch.float32,
        device=frames_u8_dev.device,
    )
    mean = (ctypes.c_float * 3)(*IV2_MEAN)
    std = (ctypes.c_float * 3)(*IV2_STD)
    hotpath.check(
        lib.cc_clip_preprocess(
            resized.data_ptr(), fnum, target_size, target_size, mean, std,
            out.data_ptr(), 0, stream,
        )
    )
    return out.unsqueeze(0)  # (1, fnum, 3, H, W) — internvideo2_mm.py:405
