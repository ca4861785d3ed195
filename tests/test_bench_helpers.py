"""bench.py helper units + aux utilities (CPU)."""

import numpy as np

import bench


def test_gemm_flops_accounting_matches_hand_calc():
    # ViT-B/32: patch 2*49*3072*768 + 12*(qkv+out+fc1+fc2) + proj
    per_layer = 2 * 50 * 768 * 2304 + 2 * 50 * 768 * 768 \
        + 2 * 50 * 768 * 3072 + 2 * 50 * 3072 * 768
    want = 2 * 49 * 3072 * 768 + 12 * per_layer + 2 * 768 * 512
    assert bench.gemm_flops_per_frame("vit_b32") == float(want)
    # L/14 uses the padded patch K (588 -> 640)
    l14 = bench.gemm_flops_per_frame("vit_l14")
    per_layer14 = 2 * 257 * 1024 * 3072 + 2 * 257 * 1024 * 1024 \
        + 2 * 257 * 1024 * 4096 + 2 * 257 * 4096 * 1024
    assert l14 == float(2 * 256 * 588 * 1024 + 24 * per_layer14 + 2 * 1024 * 768)


def test_make_nv12_batch_shape_and_determinism():
    y1, uv1 = bench.make_nv12_batch(4, seed=5)
    y2, uv2 = bench.make_nv12_batch(4, seed=5)
    assert y1.shape == (4, 1088, 1920) and uv1.shape == (4, 544, 1920)
    assert y1.dtype == np.uint8 and uv1.dtype == np.uint8
    np.testing.assert_array_equal(y1, y2)
    np.testing.assert_array_equal(uv1, uv2)
    y3, _ = bench.make_nv12_batch(4, seed=6)
    assert not np.array_equal(y1, y3)


def test_roctx_noop_without_library():
    from cosmos_curate_amd.core.utils.roctx import annotate, roctx_range

    with roctx_range("x"):
        pass

    @annotate("y")
    def f():
        return 42

    assert f() == 42


def test_hardware_info_gpu_less():
    from cosmos_curate_amd.core.utils.hardware_info import get_gpu_infos

    import torch

    infos = get_gpu_infos()
    if not torch.cuda.is_available():
        assert infos == []
    else:
        assert infos and infos[0].memory_total_mb > 0
