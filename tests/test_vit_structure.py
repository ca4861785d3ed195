"""Product ViT arithmetic vs the transformers oracle (CPU, no GPU).

Pins the *structure* of ClipVisionTowerAMD (patch-GEMM flattening, fused
QKV ordering, quick-gelu, LN placement, projection, L2 norm) against the
reference's own arithmetic (transformers CLIPVisionModelWithProjection,
models/clip.py:64-74) by monkeypatching the one contraction primitive
(_linear) to torch on CPU.  The MFMA kernel itself is pinned on the GPU in
tests/test_gpu_vit.py; this test makes a GPU failure attributable to the
kernel alone.
"""

import numpy as np
import pytest
import torch

from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights
from oracle import vit as oracle_vit
from oracle.color import clip_preprocess


@pytest.fixture(scope="module")
def weights():
    return make_clip_vit_b32_weights()


def _torch_linear(self, x, w, b, act=0, residual=None):
    y = torch.nn.functional.linear(x.float(), w.float(), b.float() if b is not None else None)
    if act == 1:
        y = y * torch.sigmoid(1.702 * y)
    if residual is not None:
        y = y + residual.float()
    return y.to(torch.bfloat16)


def test_weights_deterministic(weights):
    again = make_clip_vit_b32_weights()
    for k, v in weights.items():
        assert torch.equal(v, again[k]), k


def test_tower_l14_matches_transformers_fp32(monkeypatch):
    """ViT-L/14 geometry (the reference's own CLIP model, clip.py:33):
    patch-K padding (588->640) + 24-layer tower vs transformers fp32."""
    pytest.importorskip("transformers")
    from transformers import CLIPVisionConfig
    from transformers.models.clip.modeling_clip import CLIPVisionModelWithProjection

    from cosmos_curate_amd.models import clip_weights as cw

    weights = cw.make_clip_vit_weights(cw.VIT_L14)
    cfg = CLIPVisionConfig(
        hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        intermediate_size=4096, patch_size=14, projection_dim=768,
    )
    ref = CLIPVisionModelWithProjection(cfg)
    missing, unexpected = ref.load_state_dict(weights, strict=False)
    assert not [m for m in missing if "position_ids" not in m] and not unexpected
    ref = ref.float().eval()

    rng = np.random.default_rng(0x14)
    frames = rng.integers(0, 256, size=(1, 224, 224, 3), dtype=np.uint8)
    pixels = clip_preprocess(frames)
    with torch.no_grad():
        want = ref(pixel_values=torch.from_numpy(pixels)).image_embeds
        want = (want / torch.linalg.vector_norm(want, dim=-1, keepdim=True)).numpy()

    monkeypatch.setattr(ClipVisionTowerAMD, "_linear", _torch_linear)
    tower = ClipVisionTowerAMD(weights, cw.VIT_L14)
    got = tower(torch.from_numpy(pixels)).float().numpy()
    assert got.shape == (1, 768)
    cos = float(np.sum(want * got))
    assert cos >= 0.999, cos


def test_tower_matches_transformers_fp32(weights, monkeypatch):
    transformers = pytest.importorskip("transformers")  # noqa: F841
    ref = oracle_vit.build_reference_clip_vision(weights)

    rng = np.random.default_rng(0xC11F)
    frames = rng.integers(0, 256, size=(2, 224, 224, 3), dtype=np.uint8)
    pixels = clip_preprocess(frames)  # (2,3,224,224) f32

    want = oracle_vit.embed_frames_fp32(ref, pixels)

    monkeypatch.setattr(ClipVisionTowerAMD, "_linear", _torch_linear)
    tower = ClipVisionTowerAMD(weights)
    got = tower(torch.from_numpy(pixels)).float().numpy()

    assert want.shape == got.shape == (2, 512)
    cos = np.sum(want * got, axis=1)  # both unit-norm
    assert np.all(cos >= 0.999), f"cosine too low: {cos}"
    # unit norm
    np.testing.assert_allclose(np.linalg.norm(got, axis=1), 1.0, atol=1e-3)


def test_siglip_weights_load_into_transformers():
    """make_siglip_weights keys match SiglipVisionModel exactly and the
    fp32 oracle produces unit-norm pooled embeddings (tiny geometry)."""
    import torch

    from cosmos_curate_amd.models.clip_weights import VitConfig, make_siglip_weights
    from oracle.vit import build_reference_siglip_vision, siglip_embed_frames_fp32

    tiny = VitConfig("siglip_l16_256", hidden=128, layers=2, heads=2,
                     intermediate=256, patch=16, proj=128, image=32,
                     has_cls=False, act="gelu_tanh")
    sd = make_siglip_weights(tiny)
    ref = build_reference_siglip_vision(
        sd, dict(hidden_size=128, intermediate_size=256, num_hidden_layers=2,
                 num_attention_heads=2, image_size=32, patch_size=16))
    e = siglip_embed_frames_fp32(ref, torch.randn(2, 3, 32, 32))
    assert e.shape == (2, 128)
    import numpy as np

    np.testing.assert_allclose((e ** 2).sum(axis=1), 1.0, rtol=1e-5)


def test_siglip_config_geometry():
    from cosmos_curate_amd.models.clip_weights import SIGLIP_L16_256

    cfg = SIGLIP_L16_256
    assert cfg.num_pos == 256  # no CLS token
    assert cfg.hidden // cfg.heads == 64  # attn_mid head-dim contract
    assert (3 * cfg.patch * cfg.patch) % 64 == 0
