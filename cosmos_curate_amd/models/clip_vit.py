"""CLIP ViT-B/32 vision tower, MI355X-native forward.

Product counterpart of ``CLIPModel.get_image_features``
(/root/reference/cosmos_curate/models/clip.py:64-74): same arithmetic as
transformers' CLIPVisionModelWithProjection (conv patch embed -> cls+pos ->
pre-LN -> 12 x [LN1, MHA, LN2, quick-gelu MLP] -> post-LN on cls ->
visual projection), organized the MI355X way:

- every big contraction (patch-embed-as-GEMM, fused QKV, out-proj,
  fc1/fc2, visual projection) runs on the hand-written MFMA bf16 kernel
  (cc_gemm_bf16, csrc/cc_gemm.hip) through the C ABI;
- attention softmax/bmm and layernorms stay on torch-rocm
  (north_star: "the rest stays torch-rocm"); layernorms compute in f32;
- activations are bf16 end-to-end; the final embedding is cast to f32 and
  L2-normalized (clip.py:74).

No CPU fallback: _linear() requires the HIP extension + a GPU (the tests
may monkeypatch _linear to torch on CPU to pin arithmetic structure; the
product never does).
"""

from __future__ import annotations

import ctypes
import math

import torch

from cosmos_curate_amd import hotpath
from cosmos_curate_amd.models import clip_weights as cw


def _cc_linear(
    x: torch.Tensor,
    w_bf16: torch.Tensor,
    bias_f32: torch.Tensor | None,
    act: int = 0,
    residual: torch.Tensor | None = None,
) -> torch.Tensor:
    """C[M,N] = act(x[M,K] @ w[N,K]^T + bias) [+ residual], bf16 out.

    act 1 = quick-gelu; residual fused into the epilogue (cc_gemm_bf16_ex)."""
    lib = hotpath.require_gpu()
    assert x.dtype == torch.bfloat16 and w_bf16.dtype == torch.bfloat16
    x = x.contiguous()
    M, K = x.shape
    N = w_bf16.shape[0]
    out = torch.empty((M, N), dtype=torch.bfloat16, device=x.device)
    stream = torch.cuda.current_stream(x.device).cuda_stream
    if residual is not None:
        residual = residual.contiguous()
        assert residual.shape == out.shape and residual.dtype == torch.bfloat16
    hotpath.check(
        lib.cc_gemm_bf16_ex(
            x.data_ptr(), w_bf16.data_ptr(), out.data_ptr(), M, N, K,
            bias_f32.data_ptr() if bias_f32 is not None else None,
            1, act, residual.data_ptr() if residual is not None else None,
            stream,
        )
    )
    return out


class ClipVisionTowerAMD(torch.nn.Module):
    """CLIP vision tower + projection on the MFMA GEMM path.

    Parameterized over the tower geometry (clip_weights.VitConfig):
    ViT-B/32 (flagship, configs #1/#2) and ViT-L/14 (the reference's own
    CLIP model, models/clip.py:33; BASELINE config #3 class).  The
    patch-embed GEMM's K = 3*patch^2 is zero-padded to the kernel's
    64-multiple requirement (588 -> 640 for L/14) — exact, the padded
    columns multiply zeros.
    """

    def __init__(
        self,
        state_dict: dict[str, torch.Tensor] | None = None,
        cfg: cw.VitConfig = cw.VIT_B32,
    ) -> None:
        super().__init__()
        self.cfg = cfg
        if state_dict is None:
            state_dict = (
                cw.make_clip_vit_b32_weights() if cfg is cw.VIT_B32
                else cw.make_clip_vit_weights(cfg)
            )
        sd = state_dict
        p = "vision_model."
        reg = self.register_buffer
        to_bf = lambda t: t.to(torch.bfloat16).contiguous()  # noqa: E731

        k0 = 3 * cfg.patch * cfg.patch
        self.patch_k = (k0 + 63) // 64 * 64  # cc_gemm needs K % 64 == 0
        reg("cls_emb", sd[p + "embeddings.class_embedding"].clone())
        w_patch = sd[p + "embeddings.patch_embedding.weight"].reshape(cfg.hidden, k0)
        if self.patch_k != k0:
            w_patch = torch.nn.functional.pad(w_patch, (0, self.patch_k - k0))
        reg("w_patch", to_bf(w_patch))
        reg("pos_emb", sd[p + "embeddings.position_embedding.weight"].clone())
        reg("pre_ln_w", sd[p + "pre_layrnorm.weight"].clone())
        reg("pre_ln_b", sd[p + "pre_layrnorm.bias"].clone())
        self.layers = cfg.layers
        for i in range(self.layers):
            q = f"{p}encoder.layers.{i}."
            # fused QKV: [3*768, 768] rows ordered (q, k, v)
            wq, wk, wv = (sd[q + f"self_attn.{x}.weight"] for x in ("q_proj", "k_proj", "v_proj"))
            bq, bk, bv = (sd[q + f"self_attn.{x}.bias"] for x in ("q_proj", "k_proj", "v_proj"))
            reg(f"w_qkv_{i}", to_bf(torch.cat([wq, wk, wv], dim=0)))
            reg(f"b_qkv_{i}", torch.cat([bq, bk, bv]).float().contiguous())
            reg(f"w_out_{i}", to_bf(sd[q + "self_attn.out_proj.weight"]))
            reg(f"b_out_{i}", sd[q + "self_attn.out_proj.bias"].float().contiguous())
            reg(f"ln1_w_{i}", sd[q + "layer_norm1.weight"].clone())
            reg(f"ln1_b_{i}", sd[q + "layer_norm1.bias"].clone())
            reg(f"ln2_w_{i}", sd[q + "layer_norm2.weight"].clone())
            reg(f"ln2_b_{i}", sd[q + "layer_norm2.bias"].clone())
            reg(f"w_fc1_{i}", to_bf(sd[q + "mlp.fc1.weight"]))
            reg(f"b_fc1_{i}", sd[q + "mlp.fc1.bias"].float().contiguous())
            reg(f"w_fc2_{i}", to_bf(sd[q + "mlp.fc2.weight"]))
            reg(f"b_fc2_{i}", sd[q + "mlp.fc2.bias"].float().contiguous())
        reg("post_ln_w", sd[p + "post_layernorm.weight"].clone())
        reg("post_ln_b", sd[p + "post_layernorm.bias"].clone())
        reg("w_proj", to_bf(sd["visual_projection.weight"]))
        self.heads = cfg.heads
        self.scale = 1.0 / math.sqrt(cfg.hidden // cfg.heads)

    # the one contraction primitive; tests may monkeypatch this
    def _linear(
        self,
        x: torch.Tensor,
        w: torch.Tensor,
        b: torch.Tensor | None,
        act: int = 0,
        residual: torch.Tensor | None = None,
    ) -> torch.Tensor:
        return _cc_linear(x, w, b, act, residual)

    def _ln(self, x: torch.Tensor, w: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.dtype == torch.bfloat16:
            # fused HIP LN (f32 stats, torch layer_norm opmath semantics)
            lib = hotpath.require_gpu()
            xc = x.contiguous()
            out = torch.empty_like(xc)
            h = xc.shape[-1]
            m = xc.numel() // h
            stream = torch.cuda.current_stream(x.device).cuda_stream
            hotpath.check(
                lib.cc_layernorm_bf16(
                    xc.data_ptr(), w.data_ptr(), b.data_ptr(), out.data_ptr(),
                    m, h, 1e-5, stream,
                )
            )
            return out
        # CPU tensors only (the monkeypatched structure tests)
        return torch.nn.functional.layer_norm(
            x.float(), (x.shape[-1],), w, b, eps=1e-5
        ).to(x.dtype)

    @torch.no_grad()
    def forward(
        self,
        pixel_values: torch.Tensor | None = None,
        *,
        patches: torch.Tensor | None = None,
        n: int | None = None,
    ) -> torch.Tensor:
        """(N,3,224,224) bf16 -> (N,proj) f32 L2-normalized embeddings.

        Alternatively accepts prebuilt GEMM-ready ``patches``
        [N*g*g, patch_k] bf16 (cc_clip_preprocess_patches output) with
        ``n`` = N, skipping the torch reshape/permute/pad chain.
        """
        cfg = self.cfg
        g = cfg.image // cfg.patch
        k0 = 3 * cfg.patch * cfg.patch
        if patches is None:
            x = pixel_values.to(torch.bfloat16)
            n = x.shape[0]
            # patch extraction: (n,c,ph,ky,pw,kx) -> (n,ph,pw,c,ky,kx)
            patches = (
                x.reshape(n, 3, g, cfg.patch, g, cfg.patch)
                .permute(0, 2, 4, 1, 3, 5)
                .reshape(n * g * g, k0)
            )
            if self.patch_k != k0:
                patches = torch.nn.functional.pad(patches, (0, self.patch_k - k0))
        else:
            assert n is not None and patches.shape == (n * g * g, self.patch_k)
        tok_flat = self._linear(patches, self.w_patch, None)  # (n*g*g, hidden)
        seq = g * g + 1
        if tok_flat.is_cuda:
            # fused [CLS; tok] + pos-embed + pre-LN (csrc/cc_ln.hip) — one
            # bandwidth pass instead of the cat/f32-add/cast/LN chain
            lib = hotpath.require_gpu()
            h = torch.empty((n * seq, cfg.hidden), dtype=torch.bfloat16,
                            device=tok_flat.device)
            stream = torch.cuda.current_stream(tok_flat.device).cuda_stream
            hotpath.check(lib.cc_embed_assemble_ln(
                tok_flat.data_ptr(), self.cls_emb.data_ptr(),
                self.pos_emb.data_ptr(), self.pre_ln_w.data_ptr(),
                self.pre_ln_b.data_ptr(), h.data_ptr(), n, seq, cfg.hidden,
                ctypes.c_float(1e-5), stream))
            h = h.reshape(n, seq, cfg.hidden)
        else:
            tok = tok_flat.reshape(n, g * g, cfg.hidden)
            cls = self.cls_emb.to(tok.dtype).expand(n, 1, cfg.hidden)
            h = torch.cat([cls, tok], dim=1)  # (n, tokens, hidden)
            h = (h.float() + self.pos_emb.unsqueeze(0)).to(torch.bfloat16)
            h = self._ln(h, self.pre_ln_w, self.pre_ln_b)
        hd = cfg.hidden // self.heads
        for i in range(self.layers):
            res = h
            y = self._ln(h, getattr(self, f"ln1_w_{i}"), getattr(self, f"ln1_b_{i}"))
            qkv_flat = self._linear(
                y.reshape(n * seq, cfg.hidden),
                getattr(self, f"w_qkv_{i}"),
                getattr(self, f"b_qkv_{i}"),
            )
            if qkv_flat.is_cuda and hd == 64:
                # fused LDS-resident attention straight off the QKV GEMM
                # output (csrc/cc_attn.hip) — no permute copies
                lib = hotpath.require_gpu()
                attn = torch.empty(
                    (n * seq, cfg.hidden), dtype=torch.bfloat16,
                    device=qkv_flat.device,
                )
                stream = torch.cuda.current_stream(qkv_flat.device).cuda_stream
                attn_fn = (lib.cc_attn_small if seq <= 64 else
                           lib.cc_attn_mid if seq <= 288 else
                           lib.cc_attn_flash)
                hotpath.check(
                    attn_fn(
                        qkv_flat.data_ptr(), attn.data_ptr(), n, seq,
                        self.heads, cfg.hidden, ctypes.c_float(self.scale),
                        stream,
                    )
                )
            else:
                qkv = qkv_flat.reshape(n, seq, 3, self.heads, hd)
                q = qkv[:, :, 0].permute(0, 2, 1, 3)  # (n, heads, seq, hd)
                k = qkv[:, :, 1].permute(0, 2, 1, 3)
                v = qkv[:, :, 2].permute(0, 2, 1, 3)
                attn = torch.nn.functional.scaled_dot_product_attention(
                    q, k, v, scale=self.scale
                )
                attn = attn.permute(0, 2, 1, 3).reshape(n * seq, cfg.hidden)
            # residual add fused into the out-proj epilogue
            h = self._linear(
                attn, getattr(self, f"w_out_{i}"), getattr(self, f"b_out_{i}"),
                residual=res.reshape(n * seq, cfg.hidden),
            ).reshape(n, seq, cfg.hidden)

            res = h
            y = self._ln(h, getattr(self, f"ln2_w_{i}"), getattr(self, f"ln2_b_{i}"))
            # quick-gelu fused into the fc1 epilogue (transformers CLIP act)
            y = self._linear(
                y.reshape(n * seq, cfg.hidden),
                getattr(self, f"w_fc1_{i}"),
                getattr(self, f"b_fc1_{i}"),
                act=1,
            )
            h = self._linear(
                y, getattr(self, f"w_fc2_{i}"), getattr(self, f"b_fc2_{i}"),
                residual=res.reshape(n * seq, cfg.hidden),
            ).reshape(n, seq, cfg.hidden)

        pooled = self._ln(h[:, 0], self.post_ln_w, self.post_ln_b)
        emb = self._linear(pooled.to(torch.bfloat16), self.w_proj, None).float()
        return emb / torch.linalg.vector_norm(emb, dim=-1, keepdim=True)
