"""Pixel-path arithmetic of the hot path (CPU oracle, pure numpy).

Defines the exact arithmetic the HIP kernels in
``cosmos_curate_amd/csrc/cc_hotpath.hip`` must reproduce, covering the GPU
work the reference delegates to external libraries:

- NV12 -> RGB888:   replaces ``cvcuda.cvtcolor_into(..., YUV2RGB_NV12)``
                    (nvcodec_utils.py:178).  BT.601 limited-range, float32
                    coefficients, nearest (2x2 co-sited) chroma, round-half-up.
- bilinear resize:  replaces ``cvcuda.resize_into(LINEAR)``
                    (nvcodec_utils.py:189-194).  Pixel-center mapping
                    sx = (dx+0.5)*scale - 0.5, edge clamp, float32 weights.
- bicubic resize:   replaces the CPU fallback's cv2 INTER_CUBIC
                    (decoder_utils.py:666-670).  A=-0.75 cubic kernel,
                    replicate border, float32.
- CLIP normalize:   replaces the torchvision transform chain
                    (models/clip.py:48-62): u8 HWC -> f32 CHW,
                    (x/255 - mean)/std.  (At the benchmark config frames are
                    already 224x224 so Resize/CenterCrop are identity.)

The reference's own tests never pin CVCUDA/cv2 pixel output bit-exactly
(SURVEY.md §8c): end-to-end pixel parity is pinned via embedding cosine
>= 0.999.  Within THIS repo, oracle<->HIP parity is bit-exact
on u8 outputs: both sides use the same f32 formulas and rounding.
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt

# BT.601 limited-range YCbCr -> RGB (the constants cv2/CVCUDA's NV12 path is
# built from, expressed in f32; cvcuda uses the same ITU-R BT.601 matrix in
# Q20 fixed point).
_CY = np.float32(1.1643835)
_CVR = np.float32(1.5960267)
_CVG = np.float32(-0.8129676)
_CUG = np.float32(-0.3917623)
_CUB = np.float32(2.0172321)


def _round_u8(x: npt.NDArray[np.float32]) -> npt.NDArray[np.uint8]:
    """Round-half-up to uint8 with clamp -- the kernels use floorf(x+0.5f)."""
    return np.clip(np.floor(x + np.float32(0.5)), 0, 255).astype(np.uint8)


def nv12_to_rgb(y: npt.NDArray[np.uint8], uv: npt.NDArray[np.uint8]) -> npt.NDArray[np.uint8]:
    """NV12 (Y: HxW, UV interleaved: (H/2)x(W/2)x2) -> RGB888 HxWx3.

    Chroma is upsampled nearest: the 2x2 luma block (2i:2i+2, 2j:2j+2) shares
    chroma sample (i, j) -- NV12's native co-siting, as CVCUDA's
    YUV2RGB_NV12 does.
    """
    h, w = y.shape
    assert h % 2 == 0 and w % 2 == 0, (h, w)
    assert uv.shape == (h // 2, w // 2, 2), uv.shape
    yf = y.astype(np.float32) - np.float32(16.0)
    u = uv[:, :, 0].astype(np.float32) - np.float32(128.0)
    v = uv[:, :, 1].astype(np.float32) - np.float32(128.0)
    # nearest chroma upsample to full res
    u_full = np.repeat(np.repeat(u, 2, axis=0), 2, axis=1)
    v_full = np.repeat(np.repeat(v, 2, axis=0), 2, axis=1)
    fy = _CY * yf
    r = fy + _CVR * v_full
    g = fy + _CVG * v_full + _CUG * u_full
    b = fy + _CUB * u_full
    return np.stack([_round_u8(r), _round_u8(g), _round_u8(b)], axis=-1)


def _src_grid(dst_n: int, src_n: int) -> npt.NDArray[np.float32]:
    """Pixel-center source coordinates for a dst axis of length dst_n."""
    scale = np.float32(src_n) / np.float32(dst_n)
    d = np.arange(dst_n, dtype=np.float32)
    return (d + np.float32(0.5)) * scale - np.float32(0.5)


def resize_bilinear_u8(img: npt.NDArray[np.uint8], out_h: int, out_w: int) -> npt.NDArray[np.uint8]:
    """Bilinear resize of HxWxC u8, pixel-center mapping, edge clamp, f32 math."""
    h, w = img.shape[:2]
    sy = _src_grid(out_h, h)
    sx = _src_grid(out_w, w)
    y0 = np.clip(np.floor(sy), 0, h - 1).astype(np.int64)
    x0 = np.clip(np.floor(sx), 0, w - 1).astype(np.int64)
    y1 = np.minimum(y0 + 1, h - 1)
    x1 = np.minimum(x0 + 1, w - 1)
    wy = np.clip(sy - y0.astype(np.float32), 0.0, 1.0).astype(np.float32)
    wx = np.clip(sx - x0.astype(np.float32), 0.0, 1.0).astype(np.float32)

    im = img.astype(np.float32)
    top = im[y0][:, x0] * (1 - wx)[None, :, None] + im[y0][:, x1] * wx[None, :, None]
    bot = im[y1][:, x0] * (1 - wx)[None, :, None] + im[y1][:, x1] * wx[None, :, None]
    out = top * (1 - wy)[:, None, None] + bot * wy[:, None, None]
    return _round_u8(out.astype(np.float32))


def _cubic_weights(t: npt.NDArray[np.float32]) -> npt.NDArray[np.float32]:
    """cv2 INTER_CUBIC kernel, A = -0.75.  t is the fractional offset.

    Returns taps (len(t), 4) for source offsets {-1, 0, 1, 2} around floor(s).
    """
    A = np.float32(-0.75)
    t = t.astype(np.float32)
    w = np.empty((len(t), 4), dtype=np.float32)
    # distances of the 4 taps from the sample point: 1+t, t, 1-t, 2-t
    x0 = t + 1.0
    x1 = t
    x2 = 1.0 - t
    x3 = 2.0 - t
    # |x|<=1: (A+2)|x|^3 - (A+3)|x|^2 + 1 ; 1<|x|<2: A|x|^3 - 5A|x|^2 + 8A|x| - 4A
    w[:, 0] = ((A * x0 - 5 * A) * x0 + 8 * A) * x0 - 4 * A
    w[:, 1] = ((A + 2) * x1 - (A + 3)) * x1 * x1 + 1
    w[:, 2] = ((A + 2) * x2 - (A + 3)) * x2 * x2 + 1
    w[:, 3] = ((A * x3 - 5 * A) * x3 + 8 * A) * x3 - 4 * A
    return w


def resize_bicubic_u8(img: npt.NDArray[np.uint8], out_h: int, out_w: int) -> npt.NDArray[np.uint8]:
    """Bicubic (A=-0.75) resize of HxWxC u8, pixel-center mapping, replicate border."""
    h, w = img.shape[:2]
    sy = _src_grid(out_h, h)
    sx = _src_grid(out_w, w)
    iy = np.floor(sy).astype(np.int64)
    ix = np.floor(sx).astype(np.int64)
    wy = _cubic_weights((sy - iy).astype(np.float32))  # (out_h, 4)
    wx = _cubic_weights((sx - ix).astype(np.float32))  # (out_w, 4)

    im = img.astype(np.float32)
    # gather 4 rows x 4 cols with replicate clamp
    rows = np.clip(iy[:, None] + np.arange(-1, 3)[None, :], 0, h - 1)  # (out_h,4)
    cols = np.clip(ix[:, None] + np.arange(-1, 3)[None, :], 0, w - 1)  # (out_w,4)
    # horizontal pass, explicit left-to-right f32 adds (same association as
    # the HIP kernel, for bit-exact parity): ((t0+t1)+t2)+t3
    g = im[:, cols]  # (h, out_w, 4, C)
    wxf = wx.astype(np.float32)
    horiz = (
        (g[:, :, 0] * wxf[None, :, 0, None] + g[:, :, 1] * wxf[None, :, 1, None])
        + g[:, :, 2] * wxf[None, :, 2, None]
    ) + g[:, :, 3] * wxf[None, :, 3, None]  # (h, out_w, C) f32
    # vertical pass, same association
    v = horiz[rows]  # (out_h, 4, out_w, C)
    wyf = wy.astype(np.float32)
    out = (
        (v[:, 0] * wyf[:, 0, None, None] + v[:, 1] * wyf[:, 1, None, None])
        + v[:, 2] * wyf[:, 2, None, None]
    ) + v[:, 3] * wyf[:, 3, None, None]
    return _round_u8(out.astype(np.float32))


# CLIP normalization constants (models/clip.py:57-60)
CLIP_MEAN = np.array([0.48145466, 0.4578275, 0.40821073], dtype=np.float32)
CLIP_STD = np.array([0.26862954, 0.26130258, 0.27577711], dtype=np.float32)


def clip_preprocess(frames_u8: npt.NDArray[np.uint8]) -> npt.NDArray[np.float32]:
    """(N,H,W,3) u8 -> (N,3,H,W) f32 CLIP-normalized (models/clip.py:48-62).

    x/255 (ConvertImageDtype) then (x-mean)/std, channels-first.
    """
    x = frames_u8.astype(np.float32) / np.float32(255.0)
    x = (x - CLIP_MEAN[None, None, None, :]) / CLIP_STD[None, None, None, :]
    return np.ascontiguousarray(x.transpose(0, 3, 1, 2))
