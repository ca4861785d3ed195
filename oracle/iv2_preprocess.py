"""InternVideo2 input-frame preparation (CPU oracle).

Restates internvideo2_mm.py:386-438: temporal subsample [::len//fnum]
[:fnum], per-frame bilinear resize (cv2.resize default INTER_LINEAR,
pixel-center semantics restated in oracle/color.py), (x/255 - mean)/std
with the ImageNet constants, stack to (1, fnum, 3, H, W) f32.  The CE1
``formulate_input_frames`` (cosmos_embed1.py:113) follows the same
subsample + HF-processor resize/normalize family.
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt

from oracle.color import resize_bilinear_u8

IV2_MEAN = np.array([0.485, 0.456, 0.406], dtype=np.float32)
IV2_STD = np.array([0.229, 0.224, 0.225], dtype=np.float32)


def formulate_input_frames(
    frames_u8: npt.NDArray[np.uint8], fnum: int = 8, target_size: int = 224
) -> npt.NDArray[np.float32]:
    if len(frames_u8) < fnum:
        msg = f"Frame count {len(frames_u8)} is smaller than minimal requirement {fnum}"
        raise ValueError(msg)
    step = len(frames_u8) // fnum
    picked = frames_u8[::step][:fnum]
    if picked.shape[1:3] != (target_size, target_size):
        picked = np.stack(
            [resize_bilinear_u8(f, target_size, target_size) for f in picked]
        )
    x = (picked.astype(np.float32) / np.float32(255.0) - IV2_MEAN) / IV2_STD
    return np.ascontiguousarray(x.transpose(0, 3, 1, 2))[None]  # (1,fnum,3,H,W)
