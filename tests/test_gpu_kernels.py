"""HIP pixel-kernel parity vs the CPU oracle (MI355X only).

Bit-exact u8 contract: kernels and oracle evaluate the same f32 formulas
(oracle/color.py; csrc/cc_pixel.hip compiled -ffp-contract=off).
"""

import ctypes

import numpy as np
import pytest
import torch

from cosmos_curate_amd import hotpath
from oracle import color as ocolor
from oracle import sampling as osampling

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def lib():
    return hotpath.require_gpu()


def _dev(arr: np.ndarray) -> torch.Tensor:
    return torch.from_numpy(np.ascontiguousarray(arr)).cuda()


def rand_nv12(rng, n, h, w):
    y = rng.integers(0, 256, size=(n, h, w), dtype=np.uint8).astype(np.uint8)
    uv = rng.integers(0, 256, size=(n, h // 2, w), dtype=np.uint8).astype(np.uint8)
    return y, uv


def test_nv12_to_rgb_bitexact(lib):
    rng = np.random.default_rng(1)
    for (n, h, w) in [(2, 64, 96), (1, 1088, 1920)]:
        y, uv = rand_nv12(rng, n, h, w)
        yd, uvd = _dev(y), _dev(uv)
        out = torch.empty((n, h, w, 3), dtype=torch.uint8, device="cuda")
        hotpath.check(lib.cc_nv12_to_rgb(yd.data_ptr(), uvd.data_ptr(), n, h, w, w, out.data_ptr(), 0))
        torch.cuda.synchronize()
        got = out.cpu().numpy()
        for i in range(n):
            want = ocolor.nv12_to_rgb(y[i], uv[i].reshape(h // 2, w // 2, 2))
            np.testing.assert_array_equal(got[i], want, err_msg=f"frame {i} {h}x{w}")


def test_nv12_to_rgb_resize_fused_bitexact(lib):
    """Fused kernel == oracle convert(u8) -> bilinear, bit-exact."""
    rng = np.random.default_rng(2)
    for (h, w, th, tw) in [(1088, 1920, 224, 224), (64, 96, 32, 48), (64, 96, 128, 160)]:
        y, uv = rand_nv12(rng, 1, h, w)
        yd, uvd = _dev(y), _dev(uv)
        out = torch.empty((1, th, tw, 3), dtype=torch.uint8, device="cuda")
        hotpath.check(
            lib.cc_nv12_to_rgb_resize(yd.data_ptr(), uvd.data_ptr(), 1, h, w, w, out.data_ptr(), th, tw, 0)
        )
        torch.cuda.synchronize()
        rgb = ocolor.nv12_to_rgb(y[0], uv[0].reshape(h // 2, w // 2, 2))
        want = ocolor.resize_bilinear_u8(rgb, th, tw)
        np.testing.assert_array_equal(out[0].cpu().numpy(), want)


def test_resize_bilinear_bitexact(lib):
    rng = np.random.default_rng(3)
    for (h, w, th, tw) in [(1080, 1920, 224, 224), (27, 48, 54, 96), (224, 224, 224, 224)]:
        img = rng.integers(0, 256, size=(2, h, w, 3), dtype=np.uint8)
        d = _dev(img)
        out = torch.empty((2, th, tw, 3), dtype=torch.uint8, device="cuda")
        hotpath.check(lib.cc_resize_bilinear_u8(d.data_ptr(), 2, h, w, out.data_ptr(), th, tw, 0))
        torch.cuda.synchronize()
        got = out.cpu().numpy()
        for i in range(2):
            np.testing.assert_array_equal(got[i], ocolor.resize_bilinear_u8(img[i], th, tw))


def test_resize_bicubic_bitexact(lib):
    rng = np.random.default_rng(4)
    for (h, w, th, tw) in [(1080, 1920, 224, 224), (60, 80, 224, 224)]:
        img = rng.integers(0, 256, size=(1, h, w, 3), dtype=np.uint8)
        d = _dev(img)
        out = torch.empty((1, th, tw, 3), dtype=torch.uint8, device="cuda")
        hotpath.check(lib.cc_resize_bicubic_u8(d.data_ptr(), 1, h, w, out.data_ptr(), th, tw, 0))
        torch.cuda.synchronize()
        want = ocolor.resize_bicubic_u8(img[0], th, tw)
        got = out[0].cpu().numpy()
        np.testing.assert_array_equal(got, want)


def test_clip_preprocess_f32_bitexact(lib):
    rng = np.random.default_rng(5)
    frames = rng.integers(0, 256, size=(3, 224, 224, 3), dtype=np.uint8)
    d = _dev(frames)
    out = torch.empty((3, 3, 224, 224), dtype=torch.float32, device="cuda")
    mean = (ctypes.c_float * 3)(*ocolor.CLIP_MEAN)
    std = (ctypes.c_float * 3)(*ocolor.CLIP_STD)
    hotpath.check(lib.cc_clip_preprocess(d.data_ptr(), 3, 224, 224, mean, std, out.data_ptr(), 0, 0))
    torch.cuda.synchronize()
    want = ocolor.clip_preprocess(frames)
    np.testing.assert_array_equal(out.cpu().numpy(), want)


def test_clip_preprocess_bf16_matches_torch_cast(lib):
    rng = np.random.default_rng(6)
    frames = rng.integers(0, 256, size=(2, 224, 224, 3), dtype=np.uint8)
    d = _dev(frames)
    out = torch.empty((2, 3, 224, 224), dtype=torch.bfloat16, device="cuda")
    mean = (ctypes.c_float * 3)(*ocolor.CLIP_MEAN)
    std = (ctypes.c_float * 3)(*ocolor.CLIP_STD)
    hotpath.check(lib.cc_clip_preprocess(d.data_ptr(), 2, 224, 224, mean, std, out.data_ptr(), 1, 0))
    torch.cuda.synchronize()
    want = torch.from_numpy(ocolor.clip_preprocess(frames)).to(torch.bfloat16)
    assert torch.equal(out.cpu(), want)


def test_gather_broadcast_matches_oracle(lib):
    rng = np.random.default_rng(7)
    frames = rng.integers(0, 256, size=(6, 8, 8, 3), dtype=np.uint8)
    idx = np.array([0, 2, 5], dtype=np.int32)
    counts = np.array([2, 1, 3], dtype=np.int32)
    d = _dev(frames)
    out = torch.empty((6, 8, 8, 3), dtype=torch.uint8, device="cuda")
    hotpath.check(
        lib.cc_gather_frames_u8(
            d.data_ptr(), 6, 8 * 8 * 3,
            idx.ctypes.data_as(ctypes.c_void_p), counts.ctypes.data_as(ctypes.c_void_p),
            3, 6, out.data_ptr(), 0,
        )
    )
    torch.cuda.synchronize()
    want = osampling.broadcast_selected(frames, idx, counts)
    np.testing.assert_array_equal(out.cpu().numpy(), want)


def test_nv12_rejects_odd_dims(lib):
    buf = torch.zeros(64 * 63, dtype=torch.uint8, device="cuda")
    out = torch.zeros(64 * 64 * 3, dtype=torch.uint8, device="cuda")
    rc = lib.cc_nv12_to_rgb(buf.data_ptr(), buf.data_ptr(), 1, 63, 64, 64,
                            out.data_ptr(), 0)
    assert rc != 0
    rc = lib.cc_nv12_to_rgb_resize(buf.data_ptr(), buf.data_ptr(), 1, 64, 63, 64,
                                   out.data_ptr(), 32, 32, 0)
    assert rc != 0


def test_error_paths_loud():
    """ABI error convention: unsupported shapes return negative codes
    with a message, never silently fall back."""
    import torch

    from cosmos_curate_amd import hotpath

    lib = hotpath.require_gpu()
    a = torch.zeros(64, 3 * 256, dtype=torch.bfloat16, device="cuda")
    o = torch.zeros(64, 256, dtype=torch.bfloat16, device="cuda")
    # attn_flash refuses seq <= 288
    rc = lib.cc_attn_flash(a.data_ptr(), o.data_ptr(), 1, 64, 4, 256,
                           ctypes.c_float(0.125), 0)
    assert rc < 0 and b"attn_flash" in lib.cc_last_error()
    # attn_mid refuses seq <= 64
    rc = lib.cc_attn_mid(a.data_ptr(), o.data_ptr(), 1, 64, 4, 256,
                         ctypes.c_float(0.125), 0)
    assert rc < 0
    # preprocess_patches refuses kpad not covering 3*P*P
    frames = torch.zeros(1, 224, 224, 3, dtype=torch.uint8, device="cuda")
    out = torch.zeros(49, 64, dtype=torch.bfloat16, device="cuda")
    mean = (ctypes.c_float * 3)(0, 0, 0)
    std = (ctypes.c_float * 3)(1, 1, 1)
    rc = lib.cc_clip_preprocess_patches(frames.data_ptr(), 1, 224, 224, 32,
                                        64, mean, std, out.data_ptr(), 0)
    assert rc < 0
    # layernorm refuses unsupported H
    rc = lib.cc_layernorm_bf16(a.data_ptr(), a.data_ptr(), a.data_ptr(),
                               o.data_ptr(), 4, 192, 1e-5, 0)
    assert rc < 0
