"""SigLIP vision tower on the MFMA GEMM path (BASELINE config #3's
"SigLIP-L/14"-class embedder; geometry = google/siglip-large-patch16-256).

Architecture per transformers SiglipVisionModel (the numerics oracle used
by tests/test_gpu_vit.py::test_siglip_gpu_vs_oracle_cosine): conv patch
embed WITH bias, no class token, learned position embeddings, pre-norm
encoder layers with gelu_pytorch_tanh MLPs (GEMM epilogue ACT=2),
post_layernorm, then a MultiheadAttentionPooling "MAP" head (1 learned
probe query cross-attending over the tokens) whose pooled vector is the
embedding.  LayerNorm eps = 1e-6 (SiglipVisionConfig default; CLIP uses
1e-5).

Device path: same HIP kernels as the CLIP tower — cc_gemm_bf16_ex
(fused QKV / out+residual / fc1+tanh-gelu / fc2+residual),
cc_layernorm_bf16, cc_attn_mid (seq = 256 <= 288).  The tiny MAP head
(1 query per frame) runs in torch f32.
"""

from __future__ import annotations

import ctypes

import torch

from cosmos_curate_amd import hotpath
from cosmos_curate_amd.models import clip_weights as cw
from cosmos_curate_amd.models.clip_vit import _cc_linear


class SiglipVisionTowerAMD(torch.nn.Module):
    """SigLIP vision tower + MAP pooling head, bf16 MFMA path."""

    def __init__(self, sd: dict[str, torch.Tensor],
                 cfg: cw.VitConfig = cw.SIGLIP_L16_256) -> None:
        super().__init__()
        self.cfg = cfg
        self.heads = cfg.heads
        self.layers = cfg.layers
        self.scale = (cfg.hidden // cfg.heads) ** -0.5
        reg = self.register_buffer
        to_bf = lambda t: t.to(torch.bfloat16).contiguous()  # noqa: E731

        k0 = 3 * cfg.patch * cfg.patch
        self.patch_k = (k0 + 63) // 64 * 64
        w_patch = sd["embeddings.patch_embedding.weight"].reshape(cfg.hidden, k0)
        if self.patch_k != k0:
            w_patch = torch.nn.functional.pad(w_patch, (0, self.patch_k - k0))
        reg("w_patch", to_bf(w_patch))
        reg("b_patch", sd["embeddings.patch_embedding.bias"].float().contiguous())
        reg("pos_emb", sd["embeddings.position_embedding.weight"].clone())
        for i in range(cfg.layers):
            q = f"encoder.layers.{i}."
            wq = sd[q + "self_attn.q_proj.weight"]
            wk = sd[q + "self_attn.k_proj.weight"]
            wv = sd[q + "self_attn.v_proj.weight"]
            bq = sd[q + "self_attn.q_proj.bias"]
            bk = sd[q + "self_attn.k_proj.bias"]
            bv = sd[q + "self_attn.v_proj.bias"]
            reg(f"w_qkv_{i}", to_bf(torch.cat([wq, wk, wv], dim=0)))
            reg(f"b_qkv_{i}", torch.cat([bq, bk, bv]).float().contiguous())
            reg(f"w_out_{i}", to_bf(sd[q + "self_attn.out_proj.weight"]))
            reg(f"b_out_{i}", sd[q + "self_attn.out_proj.bias"].float().contiguous())
            reg(f"ln1_w_{i}", sd[q + "layer_norm1.weight"].float().contiguous())
            reg(f"ln1_b_{i}", sd[q + "layer_norm1.bias"].float().contiguous())
            reg(f"ln2_w_{i}", sd[q + "layer_norm2.weight"].float().contiguous())
            reg(f"ln2_b_{i}", sd[q + "layer_norm2.bias"].float().contiguous())
            reg(f"w_fc1_{i}", to_bf(sd[q + "mlp.fc1.weight"]))
            reg(f"b_fc1_{i}", sd[q + "mlp.fc1.bias"].float().contiguous())
            reg(f"w_fc2_{i}", to_bf(sd[q + "mlp.fc2.weight"]))
            reg(f"b_fc2_{i}", sd[q + "mlp.fc2.bias"].float().contiguous())
        reg("post_ln_w", sd["post_layernorm.weight"].float().contiguous())
        reg("post_ln_b", sd["post_layernorm.bias"].float().contiguous())
        # MAP head (torch f32)
        reg("probe", sd["head.probe"].float().contiguous())
        reg("h_inproj_w", sd["head.attention.in_proj_weight"].float().contiguous())
        reg("h_inproj_b", sd["head.attention.in_proj_bias"].float().contiguous())
        reg("h_outproj_w", sd["head.attention.out_proj.weight"].float().contiguous())
        reg("h_outproj_b", sd["head.attention.out_proj.bias"].float().contiguous())
        reg("h_ln_w", sd["head.layernorm.weight"].float().contiguous())
        reg("h_ln_b", sd["head.layernorm.bias"].float().contiguous())
        reg("h_fc1_w", sd["head.mlp.fc1.weight"].float().contiguous())
        reg("h_fc1_b", sd["head.mlp.fc1.bias"].float().contiguous())
        reg("h_fc2_w", sd["head.mlp.fc2.weight"].float().contiguous())
        reg("h_fc2_b", sd["head.mlp.fc2.bias"].float().contiguous())

    def _ln(self, x: torch.Tensor, w: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.dtype == torch.bfloat16:
            lib = hotpath.require_gpu()
            xc = x.contiguous()
            out = torch.empty_like(xc)
            h = xc.shape[-1]
            m = xc.numel() // h
            stream = torch.cuda.current_stream(x.device).cuda_stream
            hotpath.check(
                lib.cc_layernorm_bf16(
                    xc.data_ptr(), w.data_ptr(), b.data_ptr(), out.data_ptr(),
                    m, h, 1e-6, stream,  # SigLIP layer_norm_eps
                )
            )
            return out
        return torch.nn.functional.layer_norm(
            x.float(), (x.shape[-1],), w, b, eps=1e-6
        ).to(x.dtype)

    def _map_head(self, x: torch.Tensor) -> torch.Tensor:
        """MAP pooling head (SiglipMultiheadAttentionPoolingHead), torch
        f32: probe cross-attention over the (B, seq, H) tokens."""
        cfg = self.cfg
        B = x.shape[0]
        hd = cfg.hidden // self.heads
        xf = x.float()
        probe = self.probe.expand(B, 1, cfg.hidden)
        wq, wk, wv = self.h_inproj_w.chunk(3, dim=0)
        bq, bk, bv = self.h_inproj_b.chunk(3, dim=0)
        q = (probe @ wq.T + bq).reshape(B, 1, self.heads, hd).transpose(1, 2)
        k = (xf @ wk.T + bk).reshape(B, -1, self.heads, hd).transpose(1, 2)
        v = (xf @ wv.T + bv).reshape(B, -1, self.heads, hd).transpose(1, 2)
        att = torch.softmax((q @ k.transpose(-1, -2)) * (hd ** -0.5), dim=-1)
        o = (att @ v).transpose(1, 2).reshape(B, 1, cfg.hidden)
        o = o @ self.h_outproj_w.T + self.h_outproj_b
        res = o
        o = torch.nn.functional.layer_norm(
            o, (cfg.hidden,), self.h_ln_w, self.h_ln_b, eps=1e-6)
        o = o @ self.h_fc1_w.T + self.h_fc1_b
        o = torch.nn.functional.gelu(o, approximate="tanh")
        o = o @ self.h_fc2_w.T + self.h_fc2_b
        return (res + o)[:, 0]

    @torch.no_grad()
    def forward(
        self,
        pixel_values: torch.Tensor | None = None,
        *,
        patches: torch.Tensor | None = None,
        n: int | None = None,
    ) -> torch.Tensor:
        """(N,3,image,image) bf16 or prebuilt patches -> (N, hidden) f32
        L2-normalized MAP-pooled embeddings."""
        cfg = self.cfg
        g = cfg.image // cfg.patch
        k0 = 3 * cfg.patch * cfg.patch
        if patches is None:
            x = pixel_values.to(torch.bfloat16)
            n = x.shape[0]
            patches = (
                x.reshape(n, 3, g, cfg.patch, g, cfg.patch)
                .permute(0, 2, 4, 1, 3, 5)
                .reshape(n * g * g, k0)
            )
            if self.patch_k != k0:
                patches = torch.nn.functional.pad(patches, (0, self.patch_k - k0))
        else:
            assert n is not None and patches.shape == (n * g * g, self.patch_k)
        seq = g * g  # no class token
        tok = _cc_linear(patches, self.w_patch, self.b_patch)
        h = (tok.reshape(n, seq, cfg.hidden).float()
             + self.pos_emb.unsqueeze(0)).to(torch.bfloat16)

        hd = cfg.hidden // self.heads
        for i in range(self.layers):
            res = h
            y = self._ln(h, getattr(self, f"ln1_w_{i}"), getattr(self, f"ln1_b_{i}"))
            qkv_flat = _cc_linear(
                y.reshape(n * seq, cfg.hidden),
                getattr(self, f"w_qkv_{i}"),
                getattr(self, f"b_qkv_{i}"),
            )
            if qkv_flat.is_cuda and hd == 64:
                lib = hotpath.require_gpu()
                attn = torch.empty(
                    (n * seq, cfg.hidden), dtype=torch.bfloat16,
                    device=qkv_flat.device,
                )
                stream = torch.cuda.current_stream(qkv_flat.device).cuda_stream
                attn_fn = (lib.cc_attn_small if seq <= 64 else
                           lib.cc_attn_mid if seq <= 288 else
                           lib.cc_attn_flash)
                hotpath.check(attn_fn(
                    qkv_flat.data_ptr(), attn.data_ptr(), n, seq, self.heads,
                    cfg.hidden, ctypes.c_float(self.scale), stream,
                ))
            else:
                qkv = qkv_flat.reshape(n, seq, 3, self.heads, hd)
                q = qkv[:, :, 0].permute(0, 2, 1, 3)
                k = qkv[:, :, 1].permute(0, 2, 1, 3)
                v = qkv[:, :, 2].permute(0, 2, 1, 3)
                attn = torch.nn.functional.scaled_dot_product_attention(
                    q, k, v, scale=self.scale
                ).permute(0, 2, 1, 3).reshape(n * seq, cfg.hidden)
            h = _cc_linear(
                attn, getattr(self, f"w_out_{i}"), getattr(self, f"b_out_{i}"),
                residual=res.reshape(n * seq, cfg.hidden),
            ).reshape(n, seq, cfg.hidden)
            res = h
            y = self._ln(h, getattr(self, f"ln2_w_{i}"), getattr(self, f"ln2_b_{i}"))
            z = _cc_linear(
                y.reshape(n * seq, cfg.hidden),
                getattr(self, f"w_fc1_{i}"), getattr(self, f"b_fc1_{i}"),
                act=2,  # gelu_pytorch_tanh
            )
            h = _cc_linear(
                z, getattr(self, f"w_fc2_{i}"), getattr(self, f"b_fc2_{i}"),
                residual=res.reshape(n * seq, cfg.hidden),
            ).reshape(n, seq, cfg.hidden)

        h = self._ln(h, self.post_ln_w, self.post_ln_b)
        pooled = self._map_head(h)
        return pooled / torch.linalg.vector_norm(pooled, dim=-1, keepdim=True)
