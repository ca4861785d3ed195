"""Deterministic fixed-seed CLIP-ViT-B/32 vision weights.

HF hub weights are unavailable offline (no network in either container), so
the rebuild's stand-in for the model-weight download step
(/root/reference/cosmos_curate/core/interfaces/pipeline_interface.py:89-99)
generates weights deterministically: every tensor from its own
name-seeded torch.Generator, normal(0, 0.02) for projections and
embeddings, ones/zeros for layernorms.  Oracle (transformers fp32 CPU) and
product (MFMA bf16 GPU) load the SAME state dict, so embedding-cosine
parity (BASELINE.json >= 0.999) is meaningful.

Key names follow transformers' CLIPVisionModelWithProjection so the oracle
can load them verbatim (oracle/vit.py).
"""

from __future__ import annotations

import hashlib

import torch

# ViT-B/32 vision config (transformers CLIPVisionConfig defaults)
HIDDEN = 768
LAYERS = 12
HEADS = 12
INTERMEDIATE = 3072
PATCH = 32
IMAGE = 224
PROJ = 512
NUM_POS = (IMAGE // PATCH) ** 2 + 1  # 50


def _seed_for(name: str) -> int:
    return int.from_bytes(hashlib.sha256(name.encode()).digest()[:8], "little")


def _randn(name: str, *shape: int, std: float = 0.02) -> torch.Tensor:
    g = torch.Generator().manual_seed(_seed_for(name))
    return torch.randn(*shape, generator=g, dtype=torch.float32) * std


def make_clip_vit_b32_weights() -> dict[str, torch.Tensor]:
    """State dict for CLIPVisionModelWithProjection (ViT-B/32)."""
    sd: dict[str, torch.Tensor] = {}
    p = "vision_model."
    sd[p + "embeddings.class_embedding"] = _randn("cls", HIDDEN)
    sd[p + "embeddings.patch_embedding.weight"] = _randn(
        "patch", HIDDEN, 3, PATCH, PATCH
    )
    sd[p + "embeddings.position_embedding.weight"] = _randn("pos", NUM_POS, HIDDEN)
    # transformers' historical key spelling: pre_layrnorm
    sd[p + "pre_layrnorm.weight"] = torch.ones(HIDDEN)
    sd[p + "pre_layrnorm.bias"] = torch.zeros(HIDDEN)
    for i in range(LAYERS):
        q = f"{p}encoder.layers.{i}."
        for proj in ["q_proj", "k_proj", "v_proj", "out_proj"]:
            sd[q + f"self_attn.{proj}.weight"] = _randn(f"l{i}.{proj}.w", HIDDEN, HIDDEN)
            sd[q + f"self_attn.{proj}.bias"] = _randn(f"l{i}.{proj}.b", HIDDEN)
        sd[q + "layer_norm1.weight"] = torch.ones(HIDDEN)
        sd[q + "layer_norm1.bias"] = torch.zeros(HIDDEN)
        sd[q + "layer_norm2.weight"] = torch.ones(HIDDEN)
        sd[q + "layer_norm2.bias"] = torch.zeros(HIDDEN)
        sd[q + "mlp.fc1.weight"] = _randn(f"l{i}.fc1.w", INTERMEDIATE, HIDDEN)
        sd[q + "mlp.fc1.bias"] = _randn(f"l{i}.fc1.b", INTERMEDIATE)
        sd[q + "mlp.fc2.weight"] = _randn(f"l{i}.fc2.w", HIDDEN, INTERMEDIATE)
        sd[q + "mlp.fc2.bias"] = _randn(f"l{i}.fc2.b", HIDDEN)
    sd[p + "post_layernorm.weight"] = torch.ones(HIDDEN)
    sd[p + "post_layernorm.bias"] = torch.zeros(HIDDEN)
    sd["visual_projection.weight"] = _randn("vproj", PROJ, HIDDEN)
    return sd
