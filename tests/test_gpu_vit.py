"""MFMA GEMM + ViT parity on MI355X.

- cc_gemm_bf16 vs torch fp32 matmul of the bf16-rounded operands;
- full ClipVisionTowerAMD (GPU bf16, custom GEMMs) vs the transformers
  fp32 CPU oracle: embedding cosine >= 0.999 (BASELINE.json contract).
"""

import numpy as np
import pytest
import torch

from cosmos_curate_amd import hotpath

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def lib():
    return hotpath.require_gpu()


def cc_gemm(lib, a_bf16, b_bf16, bias=None, out_bf16=True):
    M, K = a_bf16.shape
    N = b_bf16.shape[0]
    out = torch.empty(
        (M, N), dtype=torch.bfloat16 if out_bf16 else torch.float32, device="cuda"
    )
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(
        lib.cc_gemm_bf16(
            a_bf16.contiguous().data_ptr(), b_bf16.contiguous().data_ptr(),
            out.data_ptr(), M, N, K,
            bias.contiguous().data_ptr() if bias is not None else None,
            1 if out_bf16 else 0, stream,
        )
    )
    torch.cuda.synchronize()
    return out


@pytest.mark.parametrize(
    ("M", "N", "K"),
    [
        (128, 128, 64),     # single tile
        (256, 256, 128),    # multi-tile
        (50, 512, 768),     # M, N edges (visual projection shape, M=tokens)
        (8232, 768, 3072),  # patch-embed GEMM at batch 168 frames
        (400, 2304, 768),   # fused QKV
        (33, 128, 64),      # ragged M
        (300, 1600, 256),   # 256-tile body (N trigger), ragged M+N tails
        (2000, 768, 1536),  # 256-tile body (K trigger), tail M
    ],
)
def test_gemm_vs_torch(lib, M, N, K):
    torch.manual_seed(M * 31 + N * 7 + K)
    a = (torch.randn(M, K) * 0.5).to(torch.bfloat16).cuda()
    b = (torch.randn(N, K) * 0.5).to(torch.bfloat16).cuda()
    bias = torch.randn(N).float().cuda()
    got = cc_gemm(lib, a, b, bias, out_bf16=False).cpu()
    want = a.float().cpu() @ b.float().cpu().T + bias.cpu()
    # f32 accumulate over bf16 products: tight tolerance scaled by K
    torch.testing.assert_close(got, want, rtol=5e-3, atol=5e-2)


def test_gemm_bf16_out_and_no_bias(lib):
    torch.manual_seed(0)
    a = torch.randn(130, 192).to(torch.bfloat16).cuda()
    b = torch.randn(140, 192).to(torch.bfloat16).cuda()
    got = cc_gemm(lib, a, b, None, out_bf16=True).float().cpu()
    want = (a.float().cpu() @ b.float().cpu().T)
    torch.testing.assert_close(got, want, rtol=2e-2, atol=5e-2)


@pytest.mark.parametrize(("M", "N", "K"), [(200, 256, 128), (513, 1600, 1536)])
def test_gemm_fused_gelu_and_residual(lib, M, N, K):
    torch.manual_seed(3)
    a = torch.randn(M, K).to(torch.bfloat16).cuda()
    b = torch.randn(N, K).to(torch.bfloat16).cuda()
    bias = torch.randn(N).float().cuda()
    res = torch.randn(M, N).to(torch.bfloat16).cuda()
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(
        lib.cc_gemm_bf16_ex(a.data_ptr(), b.data_ptr(), out.data_ptr(), M, N, K,
                            bias.data_ptr(), 1, 1, res.data_ptr(), stream)
    )
    torch.cuda.synchronize()
    y = a.float().cpu() @ b.float().cpu().T + bias.cpu()
    y = y * torch.sigmoid(1.702 * y) + res.float().cpu()
    torch.testing.assert_close(out.float().cpu(), y, rtol=2e-2, atol=8e-2)


def test_gemm_rejects_bad_k(lib):
    a = torch.randn(16, 60).to(torch.bfloat16).cuda()
    b = torch.randn(16, 60).to(torch.bfloat16).cuda()
    out = torch.empty((16, 16), dtype=torch.float32, device="cuda")
    rc = lib.cc_gemm_bf16(a.data_ptr(), b.data_ptr(), out.data_ptr(), 16, 16, 60, None, 0, 0)
    assert rc != 0  # K % 64 != 0 must fail loudly


def test_vit_gpu_vs_oracle_cosine(lib):
    from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
    from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights
    from oracle import vit as oracle_vit
    from oracle.color import clip_preprocess

    weights = make_clip_vit_b32_weights()
    rng = np.random.default_rng(0xBEEF)
    frames = rng.integers(0, 256, size=(4, 224, 224, 3), dtype=np.uint8)
    pixels = clip_preprocess(frames)

    ref = oracle_vit.build_reference_clip_vision(weights)
    want = oracle_vit.embed_frames_fp32(ref, pixels)

    tower = ClipVisionTowerAMD(weights).cuda()
    got = tower(torch.from_numpy(pixels).cuda()).cpu().numpy()

    cos = np.sum(want * got, axis=1)
    assert np.all(cos >= 0.999), f"embedding cosine vs fp32 oracle: {cos}"


@pytest.mark.parametrize("n,seq,heads", [(3, 50, 12), (1, 1, 2), (5, 64, 4), (2, 33, 8)])
def test_attn_small_vs_torch_sdpa(lib, n, seq, heads):
    import ctypes

    import math

    hd = 64
    hidden = heads * hd
    torch.manual_seed(n * 100 + seq)
    qkv = torch.randn(n * seq, 3 * hidden).to(torch.bfloat16).cuda()
    out = torch.empty((n * seq, hidden), dtype=torch.bfloat16, device="cuda")
    scale = 1.0 / math.sqrt(hd)
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(
        lib.cc_attn_small(qkv.data_ptr(), out.data_ptr(), n, seq, heads,
                          hidden, ctypes.c_float(scale), stream)
    )
    torch.cuda.synchronize()
    q, k, v = (
        qkv.reshape(n, seq, 3, heads, hd)[:, :, i].permute(0, 2, 1, 3).float()
        for i in range(3)
    )
    want = torch.nn.functional.scaled_dot_product_attention(q, k, v, scale=scale)
    want = want.permute(0, 2, 1, 3).reshape(n * seq, hidden)
    torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("M,H", [(1, 128), (333, 768), (1024, 1024), (50, 4096)])
def test_layernorm_kernel_vs_torch(lib, M, H):
    torch.manual_seed(M + H)
    x = (torch.randn(M, H) * 2).to(torch.bfloat16).cuda()
    w = torch.randn(H).float().cuda()
    b = torch.randn(H).float().cuda()
    out = torch.empty_like(x)
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(
        lib.cc_layernorm_bf16(x.data_ptr(), w.data_ptr(), b.data_ptr(),
                              out.data_ptr(), M, H, ctypes_float(1e-5), stream)
    )
    torch.cuda.synchronize()
    want = torch.nn.functional.layer_norm(x.float(), (H,), w, b, eps=1e-5).to(
        torch.bfloat16
    )
    torch.testing.assert_close(out, want, rtol=2e-2, atol=2e-2)
    # element-exact in most positions (same f32 math up to reduction order)
    frac_diff = (out != want).float().mean().item()
    assert frac_diff < 0.02, frac_diff


def ctypes_float(v):
    import ctypes

    return ctypes.c_float(v)


def test_vit_l14_gpu_vs_oracle_cosine(lib):
    """ViT-L/14 (reference's CLIP model geometry) on the MFMA path."""
    from transformers import CLIPVisionConfig
    from transformers.models.clip.modeling_clip import CLIPVisionModelWithProjection

    from cosmos_curate_amd.models import clip_weights as cw
    from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
    from oracle.color import clip_preprocess

    weights = cw.make_clip_vit_weights(cw.VIT_L14)
    cfg = CLIPVisionConfig(
        hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
        intermediate_size=4096, patch_size=14, projection_dim=768,
    )
    ref = CLIPVisionModelWithProjection(cfg)
    ref.load_state_dict(weights, strict=False)
    ref = ref.float().eval()

    rng = np.random.default_rng(0x114)
    frames = rng.integers(0, 256, size=(2, 224, 224, 3), dtype=np.uint8)
    pixels = clip_preprocess(frames)
    with torch.no_grad():
        want = ref(pixel_values=torch.from_numpy(pixels)).image_embeds
        want = (want / torch.linalg.vector_norm(want, dim=-1, keepdim=True)).numpy()

    tower = ClipVisionTowerAMD(weights, cw.VIT_L14).cuda()
    got = tower(torch.from_numpy(pixels).cuda()).cpu().numpy()
    cos = np.sum(want * got, axis=1)
    assert np.all(cos >= 0.999), cos


def test_clip_model_interface_end_to_end(lib):
    """CLIPImageEmbeddings from u8 frames (models/clip.py:108-118 surface)."""
    from cosmos_curate_amd.models.clip import CLIPImageEmbeddings
    from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights
    from oracle import vit as oracle_vit
    from oracle.color import clip_preprocess

    model = CLIPImageEmbeddings()
    assert model.model_id_names == ["openai/clip-vit-base-patch32"]
    model.setup()
    rng = np.random.default_rng(1)
    frames = rng.integers(0, 256, size=(3, 224, 224, 3), dtype=np.uint8)
    emb = model(frames)
    assert emb.shape == (3, 512)
    norms = torch.linalg.vector_norm(emb, dim=-1).cpu()
    torch.testing.assert_close(norms, torch.ones(3), rtol=1e-3, atol=1e-3)

    ref = oracle_vit.build_reference_clip_vision(make_clip_vit_b32_weights())
    want = oracle_vit.embed_frames_fp32(ref, clip_preprocess(frames))
    cos = np.sum(want * emb.cpu().numpy(), axis=1)
    assert np.all(cos >= 0.999), cos


def test_iv2_preprocess_matches_oracle(lib):
    """IV2 input-frame math (subsample/resize/normalize) — SURVEY.md §2
    'Embedding models' preprocessing-parity contract."""
    from cosmos_curate_amd.models import internvideo2_prep as prep
    from oracle import iv2_preprocess as oprep

    rng = np.random.default_rng(0x12)
    frames = rng.integers(0, 256, size=(21, 256, 320, 3), dtype=np.uint8)
    want = oprep.formulate_input_frames(frames, fnum=8, target_size=224)
    got = prep.formulate_input_frames(
        torch.from_numpy(frames).cuda(), fnum=8, target_size=224
    ).cpu().numpy()
    assert got.shape == want.shape == (1, 8, 3, 224, 224)
    np.testing.assert_array_equal(got, want)  # bit-exact: same f32 chain

    # subsample rule parity at awkward lengths
    from cosmos_curate_amd.models.internvideo2_prep import temporal_subsample

    for t in [8, 9, 15, 21, 100]:
        np.testing.assert_array_equal(
            temporal_subsample(t, 8), np.arange(t)[:: t // 8][:8]
        )
    with pytest.raises(ValueError):
        temporal_subsample(5, 8)


def test_embed_assemble_ln_vs_torch(lib):
    """cc_embed_assemble_ln == cat + f32 pos-add + bf16 cast + LN chain."""
    import ctypes

    torch.manual_seed(17)
    n, tokens, H = 7, 50, 768
    tok = torch.randn(n * (tokens - 1), H).to(torch.bfloat16).cuda()
    cls = torch.randn(H).float().cuda()
    pos = torch.randn(tokens, H).float().cuda()
    w = torch.randn(H).float().cuda()
    b = torch.randn(H).float().cuda()
    out = torch.empty(n * tokens, H, dtype=torch.bfloat16, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(lib.cc_embed_assemble_ln(
        tok.data_ptr(), cls.data_ptr(), pos.data_ptr(), w.data_ptr(),
        b.data_ptr(), out.data_ptr(), n, tokens, H, ctypes.c_float(1e-5),
        stream))
    torch.cuda.synchronize()

    h = torch.cat([cls.to(torch.bfloat16).expand(n, 1, H),
                   tok.reshape(n, tokens - 1, H)], dim=1)
    h = (h.float() + pos.unsqueeze(0)).to(torch.bfloat16)
    want = torch.nn.functional.layer_norm(
        h.float(), (H,), w, b, eps=1e-5).reshape(n * tokens, H)
    torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2)


def test_attn_mid_vs_sdpa(lib):
    """cc_attn_mid (seq=257, L/14 shape) == torch sdpa on the same QKV."""
    import ctypes

    torch.manual_seed(23)
    n, seq, heads, hd = 3, 257, 16, 64
    hidden = heads * hd
    qkv = (torch.randn(n * seq, 3 * hidden) * 0.5).to(torch.bfloat16).cuda()
    out = torch.empty(n * seq, hidden, dtype=torch.bfloat16, device="cuda")
    scale = hd ** -0.5
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(lib.cc_attn_mid(
        qkv.data_ptr(), out.data_ptr(), n, seq, heads, hidden,
        ctypes.c_float(scale), stream))
    torch.cuda.synchronize()

    q3 = qkv.reshape(n, seq, 3, heads, hd)
    q = q3[:, :, 0].permute(0, 2, 1, 3).float()
    k = q3[:, :, 1].permute(0, 2, 1, 3).float()
    v = q3[:, :, 2].permute(0, 2, 1, 3).float()
    want = torch.nn.functional.scaled_dot_product_attention(q, k, v, scale=scale)
    want = want.permute(0, 2, 1, 3).reshape(n * seq, hidden)
    torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2)


def test_attn_mid_ragged_seqs(lib):
    """Mask correctness at non-multiple-of-64 sequence lengths."""
    import ctypes

    for seq in [65, 100, 288]:
        torch.manual_seed(seq)
        n, heads, hd = 2, 4, 64
        hidden = heads * hd
        qkv = (torch.randn(n * seq, 3 * hidden) * 0.5).to(torch.bfloat16).cuda()
        out = torch.empty(n * seq, hidden, dtype=torch.bfloat16, device="cuda")
        scale = hd ** -0.5
        stream = torch.cuda.current_stream().cuda_stream
        hotpath.check(lib.cc_attn_mid(
            qkv.data_ptr(), out.data_ptr(), n, seq, heads, hidden,
            ctypes.c_float(scale), stream))
        torch.cuda.synchronize()
        q3 = qkv.reshape(n, seq, 3, heads, hd)
        q = q3[:, :, 0].permute(0, 2, 1, 3).float()
        k = q3[:, :, 1].permute(0, 2, 1, 3).float()
        v = q3[:, :, 2].permute(0, 2, 1, 3).float()
        want = torch.nn.functional.scaled_dot_product_attention(q, k, v, scale=scale)
        want = want.permute(0, 2, 1, 3).reshape(n * seq, hidden)
        torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2,
                                   msg=f"seq={seq}")


def test_preprocess_patches_matches_torch_chain(lib):
    """cc_clip_preprocess_patches == cc_clip_preprocess + torch
    reshape/permute/pad, bit-exact (same normalize arithmetic)."""
    from cosmos_curate_amd.models.clip import _CLIPImageEmbeddings

    m = _CLIPImageEmbeddings("vit_b32")
    torch.manual_seed(4)
    frames = torch.randint(0, 256, (5, 224, 224, 3), dtype=torch.uint8).cuda()
    fused = m.preprocess_patches_u8(frames)

    pixels = m.preprocess_u8(frames)
    cfg = m.tower.cfg
    g = cfg.image // cfg.patch
    k0 = 3 * cfg.patch * cfg.patch
    want = (
        pixels.reshape(5, 3, g, cfg.patch, g, cfg.patch)
        .permute(0, 2, 4, 1, 3, 5)
        .reshape(5 * g * g, k0)
    )
    if m.tower.patch_k != k0:
        want = torch.nn.functional.pad(want, (0, m.tower.patch_k - k0))
    assert torch.equal(fused, want)
    # and the end-to-end u8 path equals the pixels path
    e1 = m.tower(patches=fused, n=5)
    e2 = m.tower(pixels)
    assert torch.equal(e1, e2)


def test_siglip_gpu_vs_oracle_cosine(lib):
    """SigLIP-L16-256 tower (BASELINE config #3 class) on the MFMA path
    vs transformers fp32 SiglipVisionModel: cosine >= 0.999."""
    from cosmos_curate_amd.models import clip_weights as cw
    from cosmos_curate_amd.models.siglip_vit import SiglipVisionTowerAMD
    from oracle.vit import build_reference_siglip_vision, siglip_embed_frames_fp32

    cfg = cw.SIGLIP_L16_256
    sd = cw.make_siglip_weights(cfg)
    tower = SiglipVisionTowerAMD(sd, cfg).cuda()
    torch.manual_seed(6)
    pix = torch.rand(3, 3, 256, 256) * 2 - 1  # SigLIP-normalized range
    got = tower(pix.cuda().to(torch.bfloat16)).cpu().numpy()
    ref = build_reference_siglip_vision(sd)
    want = siglip_embed_frames_fp32(ref, pix)
    cos = (got * want).sum(axis=1)
    assert cos.min() > 0.999, cos


def test_siglip_model_interface_u8_path(lib):
    """CLIPImageEmbeddings(variant='siglip_l16_256') over u8 frames: the
    fused patches path and the pixels path agree bit-exactly."""
    from cosmos_curate_amd.models.clip import _CLIPImageEmbeddings

    m = _CLIPImageEmbeddings("siglip_l16_256")
    torch.manual_seed(8)
    frames = torch.randint(0, 256, (2, 256, 256, 3), dtype=torch.uint8).cuda()
    e1 = m(frames)
    pixels = m.preprocess_u8(frames)
    e2 = m.tower(pixels)
    assert torch.equal(e1, e2)
    assert e1.shape == (2, 1024)


@pytest.mark.parametrize("seq", [576, 700, 1024])
def test_attn_flash_vs_sdpa(lib, seq):
    """cc_attn_flash (online-softmax K/V streaming) == torch sdpa at
    SigLIP-384-class and larger sequence lengths."""
    import ctypes

    torch.manual_seed(seq)
    n, heads, hd = 2, 4, 64
    hidden = heads * hd
    qkv = (torch.randn(n * seq, 3 * hidden) * 0.5).to(torch.bfloat16).cuda()
    out = torch.empty(n * seq, hidden, dtype=torch.bfloat16, device="cuda")
    scale = hd ** -0.5
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(lib.cc_attn_flash(
        qkv.data_ptr(), out.data_ptr(), n, seq, heads, hidden,
        ctypes.c_float(scale), stream))
    torch.cuda.synchronize()
    q3 = qkv.reshape(n, seq, 3, heads, hd)
    q = q3[:, :, 0].permute(0, 2, 1, 3).float()
    k = q3[:, :, 1].permute(0, 2, 1, 3).float()
    v = q3[:, :, 2].permute(0, 2, 1, 3).float()
    want = torch.nn.functional.scaled_dot_product_attention(q, k, v, scale=scale)
    want = want.permute(0, 2, 1, 3).reshape(n * seq, hidden)
    torch.testing.assert_close(out.float(), want, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize(("M", "N", "K"), [(200, 256, 128), (300, 1600, 1536)])
def test_gemm_tanh_gelu_epilogue(lib, M, N, K):
    """ACT=2 (gelu_pytorch_tanh, SigLIP) epilogue on both GEMM bodies."""
    torch.manual_seed(9)
    a = torch.randn(M, K).to(torch.bfloat16).cuda()
    b = torch.randn(N, K).to(torch.bfloat16).cuda()
    bias = torch.randn(N).float().cuda()
    res = torch.randn(M, N).to(torch.bfloat16).cuda()
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(
        lib.cc_gemm_bf16_ex(a.data_ptr(), b.data_ptr(), out.data_ptr(), M, N, K,
                            bias.data_ptr(), 1, 2, res.data_ptr(), stream)
    )
    torch.cuda.synchronize()
    y = a.float().cpu() @ b.float().cpu().T + bias.cpu()
    y = torch.nn.functional.gelu(y, approximate="tanh") + res.float().cpu()
    torch.testing.assert_close(out.float().cpu(), y, rtol=2e-2, atol=8e-2)


def test_clip_non224_input_resized_on_device(lib):
    """target_res=-1 runs hand source-resolution frames to the embedder;
    __call__ must resize+center-crop on device (previously the
    u8 non-224 path hit a reshape failure and every clip errored).
    Parity: torchvision-equivalent CPU chain (bicubic shorter-side resize
    + center crop) -> fp32 oracle, cosine >= 0.999."""
    from cosmos_curate_amd.models.clip import CLIPImageEmbeddings
    from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights
    from oracle import vit as oracle_vit
    from oracle.color import clip_preprocess, resize_bicubic_u8

    model = CLIPImageEmbeddings()
    model.setup()
    rng = np.random.default_rng(7)
    frames = rng.integers(0, 256, size=(2, 240, 320, 3), dtype=np.uint8)
    emb = model(frames)
    assert emb.shape == (2, 512)
    norms = torch.linalg.vector_norm(emb, dim=-1).cpu()
    torch.testing.assert_close(norms, torch.ones(2), rtol=1e-3, atol=1e-3)

    # CPU reference of the same chain: shorter side -> 224 bicubic, center crop
    scale = 224 / 240
    rh, rw = max(224, round(240 * scale)), max(224, round(320 * scale))
    resized = np.stack([resize_bicubic_u8(f, rh, rw) for f in frames])
    top, left = (rh - 224) // 2, (rw - 224) // 2
    cropped = resized[:, top:top + 224, left:left + 224, :]
    ref = oracle_vit.build_reference_clip_vision(make_clip_vit_b32_weights())
    want = oracle_vit.embed_frames_fp32(ref, clip_preprocess(cropped))
    cos = np.sum(want * emb.cpu().numpy(), axis=1)
    assert np.all(cos >= 0.999), cos


def test_vit_stress_weights_pretrained_ranges(lib):
    """Pretrained-readiness parity: weights with
    outlier channels and realistic LN gains (clip_weights.
    make_clip_vit_weights_stress) drive activations into the ranges real
    CLIP checkpoints produce; bf16 GEMMs + exp/rcp fused epilogues must
    still hit cosine >= 0.999 vs the fp32 oracle on the SAME dict."""
    from cosmos_curate_amd.models import clip_weights as cw
    from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
    from oracle import vit as oracle_vit
    from oracle.color import clip_preprocess

    sd = cw.make_clip_vit_weights_stress(cw.VIT_B32)
    tower = ClipVisionTowerAMD(sd, cw.VIT_B32).to("cuda")
    rng = np.random.default_rng(11)
    frames = rng.integers(0, 256, size=(4, 224, 224, 3), dtype=np.uint8)
    pix = clip_preprocess(frames)
    with torch.no_grad():
        got = tower(torch.from_numpy(pix).to("cuda", torch.bfloat16))
    ref = oracle_vit.build_reference_clip_vision(sd)
    want = oracle_vit.embed_frames_fp32(ref, pix)
    # sanity: the stress dict actually produces big activations — probe
    # fc1 pre-act range through the oracle's own forward hooks is
    # overkill; assert the embedding magnitudes differ strongly from the
    # tame-weight case instead (unit-norm output, so check cosine only)
    cos = np.sum(want * got.cpu().float().numpy(), axis=1)
    assert np.all(cos >= 0.999), cos


def test_pretrained_checkpoint_path_gpu(lib, tmp_path):
    """CLIPImageEmbeddings(checkpoint_path=...) loads a real-format
    safetensors checkpoint and matches the fp32 oracle on it."""
    from safetensors.torch import save_file

    from cosmos_curate_amd.models import clip_weights as cw
    from cosmos_curate_amd.models.clip import CLIPImageEmbeddings
    from oracle import vit as oracle_vit
    from oracle.color import clip_preprocess

    sd = cw.make_clip_vit_weights_stress(cw.VIT_B32)
    save_file({k: v.contiguous() for k, v in sd.items()},
              str(tmp_path / "model.safetensors"))
    model = CLIPImageEmbeddings(checkpoint_path=str(tmp_path))
    model.setup()
    rng = np.random.default_rng(12)
    frames = rng.integers(0, 256, size=(2, 224, 224, 3), dtype=np.uint8)
    emb = model(frames)
    ref = oracle_vit.build_reference_clip_vision(sd)
    want = oracle_vit.embed_frames_fp32(ref, clip_preprocess(frames))
    cos = np.sum(want * emb.cpu().numpy(), axis=1)
    assert np.all(cos >= 0.999), cos


@pytest.mark.parametrize(
    ("M", "N", "K", "act", "res"),
    [
        (235200, 3072, 768, 1, False),   # fc1 at the bench batch (224 clips)
        (235200, 768, 3072, 0, True),    # fc2 + fused residual
        (230496, 768, 3072, 0, False),   # patch-embed grid (224*49*21)
    ],
)
def test_gemm_bench_shape_parity(lib, M, N, K, act, res):
    """Parity at the EXACT bench-batch shapes (persistent fleet, counted
    publish, staged epilogue): row-slice check vs torch fp32 of the
    bf16-rounded operands."""
    torch.manual_seed(K + M)
    a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
    b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
    bias = torch.randn(N).float().cuda()
    r = (torch.randn(M, N) * 0.3).to(torch.bfloat16).cuda() if res else None
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    hotpath.check(lib.cc_gemm_bf16_ex(
        a.data_ptr(), b.data_ptr(), out.data_ptr(), M, N, K,
        bias.data_ptr(), 1, act, r.data_ptr() if res else None, stream))
    torch.cuda.synchronize()
    # check a deterministic scatter of row blocks incl. the M tail
    for r0 in (0, M // 2, M - 64):
        y = a[r0:r0 + 64].float().cpu() @ b.float().cpu().T + bias.cpu()
        if act == 1:
            y = y * torch.sigmoid(1.702 * y)
        if res:
            y = y + r[r0:r0 + 64].float().cpu()
        torch.testing.assert_close(out[r0:r0 + 64].float().cpu(), y,
                                   rtol=2e-2, atol=8e-2)
