"""Deterministic fixed-seed CLIP vision-tower weights (ViT-B/32, ViT-L/14).

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

import dataclasses


@dataclasses.dataclass(frozen=True)
class VitConfig:
    """Vision-tower geometry (transformers CLIPVisionConfig fields)."""

    name: str
    hidden: int
    layers: int
    heads: int
    intermediate: int
    patch: int
    proj: int
    image: int = 224
    has_cls: bool = True       # SigLIP towers have no class token
    act: str = "quick_gelu"    # "quick_gelu" (CLIP) | "gelu_tanh" (SigLIP)

    @property
    def num_pos(self) -> int:
        return (self.image // self.patch) ** 2 + (1 if self.has_cls else 0)


# ViT-B/32 = transformers CLIPVisionConfig defaults (config #1/#2 embedder)
VIT_B32 = VitConfig("vit_b32", hidden=768, layers=12, heads=12,
                    intermediate=3072, patch=32, proj=512)
# ViT-L/14 = openai/clip-vit-large-patch14 geometry — the model the
# reference's CLIPImageEmbeddings actually loads (models/clip.py:33) and
# the L/14-class of BASELINE config #3
VIT_L14 = VitConfig("vit_l14", hidden=1024, layers=24, heads=16,
                    intermediate=4096, patch=14, proj=768)

# SigLIP-large geometry (google/siglip-large-patch16-256 class): no CLS
# token, tanh-gelu MLPs, attention-pooling MAP head — the SigLIP-L-class
# embedder BASELINE config #3 names.  proj == hidden (the MAP head's
# pooled vector is the embedding; no visual_projection).
SIGLIP_L16_256 = VitConfig("siglip_l16_256", hidden=1024, layers=24, heads=16,
                           intermediate=4096, patch=16, proj=1024, image=256,
                           has_cls=False, act="gelu_tanh")

CONFIGS = {"vit_b32": VIT_B32, "vit_l14": VIT_L14,
           "siglip_l16_256": SIGLIP_L16_256}


def make_siglip_weights(cfg: VitConfig = SIGLIP_L16_256) -> dict[str, torch.Tensor]:
    """State dict for transformers SiglipVisionModel of this geometry
    (key names exactly as SiglipVisionModel.named_parameters())."""
    sd: dict[str, torch.Tensor] = {}
    tag = cfg.name + ":"
    sd["embeddings.patch_embedding.weight"] = _randn(
        tag + "patch", cfg.hidden, 3, cfg.patch, cfg.patch)
    sd["embeddings.patch_embedding.bias"] = _randn(tag + "patch.b", cfg.hidden)
    sd["embeddings.position_embedding.weight"] = _randn(
        tag + "pos", cfg.num_pos, cfg.hidden)
    for i in range(cfg.layers):
        q = f"encoder.layers.{i}."
        for proj in ["q_proj", "k_proj", "v_proj", "out_proj"]:
            sd[q + f"self_attn.{proj}.weight"] = _randn(
                f"{tag}l{i}.{proj}.w", cfg.hidden, cfg.hidden)
            sd[q + f"self_attn.{proj}.bias"] = _randn(f"{tag}l{i}.{proj}.b", cfg.hidden)
        sd[q + "layer_norm1.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm1.bias"] = torch.zeros(cfg.hidden)
        sd[q + "layer_norm2.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm2.bias"] = torch.zeros(cfg.hidden)
        sd[q + "mlp.fc1.weight"] = _randn(f"{tag}l{i}.fc1.w", cfg.intermediate, cfg.hidden)
        sd[q + "mlp.fc1.bias"] = _randn(f"{tag}l{i}.fc1.b", cfg.intermediate)
        sd[q + "mlp.fc2.weight"] = _randn(f"{tag}l{i}.fc2.w", cfg.hidden, cfg.intermediate)
        sd[q + "mlp.fc2.bias"] = _randn(f"{tag}l{i}.fc2.b", cfg.hidden)
    sd["post_layernorm.weight"] = torch.ones(cfg.hidden)
    sd["post_layernorm.bias"] = torch.zeros(cfg.hidden)
    sd["head.probe"] = _randn(tag + "probe", 1, 1, cfg.hidden)
    sd["head.attention.in_proj_weight"] = _randn(
        tag + "head.inproj.w", 3 * cfg.hidden, cfg.hidden)
    sd["head.attention.in_proj_bias"] = _randn(tag + "head.inproj.b", 3 * cfg.hidden)
    sd["head.attention.out_proj.weight"] = _randn(
        tag + "head.outproj.w", cfg.hidden, cfg.hidden)
    sd["head.attention.out_proj.bias"] = _randn(tag + "head.outproj.b", cfg.hidden)
    sd["head.layernorm.weight"] = torch.ones(cfg.hidden)
    sd["head.layernorm.bias"] = torch.zeros(cfg.hidden)
    sd["head.mlp.fc1.weight"] = _randn(tag + "head.fc1.w", cfg.intermediate, cfg.hidden)
    sd["head.mlp.fc1.bias"] = _randn(tag + "head.fc1.b", cfg.intermediate)
    sd["head.mlp.fc2.weight"] = _randn(tag + "head.fc2.w", cfg.hidden, cfg.intermediate)
    sd["head.mlp.fc2.bias"] = _randn(tag + "head.fc2.b", cfg.hidden)
    return sd

# module-level aliases for the flagship config (bench flop accounting)
HIDDEN, LAYERS, HEADS = VIT_B32.hidden, VIT_B32.layers, VIT_B32.heads
INTERMEDIATE, PATCH, IMAGE, PROJ = (
    VIT_B32.intermediate, VIT_B32.patch, VIT_B32.image, VIT_B32.proj,
)
NUM_POS = VIT_B32.num_pos


def _seed_for(name: str) -> int:
    return int.from_bytes(hashlib.sha256(name.encode()).digest()[:8], "little")


def _randn(name: str, *shape: int, std: float = 0.02) -> torch.Tensor:
    g = torch.Generator().manual_seed(_seed_for(name))
    return torch.randn(*shape, generator=g, dtype=torch.float32) * std


def make_clip_vit_weights(cfg: VitConfig = VIT_B32) -> dict[str, torch.Tensor]:
    """State dict for CLIPVisionModelWithProjection of the given geometry."""
    sd: dict[str, torch.Tensor] = {}
    p = "vision_model."
    tag = cfg.name + ":"
    sd[p + "embeddings.class_embedding"] = _randn(tag + "cls", cfg.hidden)
    sd[p + "embeddings.patch_embedding.weight"] = _randn(
        tag + "patch", cfg.hidden, 3, cfg.patch, cfg.patch
    )
    sd[p + "embeddings.position_embedding.weight"] = _randn(
        tag + "pos", cfg.num_pos, cfg.hidden
    )
    # transformers' historical key spelling: pre_layrnorm
    sd[p + "pre_layrnorm.weight"] = torch.ones(cfg.hidden)
    sd[p + "pre_layrnorm.bias"] = torch.zeros(cfg.hidden)
    for i in range(cfg.layers):
        q = f"{p}encoder.layers.{i}."
        for proj in ["q_proj", "k_proj", "v_proj", "out_proj"]:
            sd[q + f"self_attn.{proj}.weight"] = _randn(
                f"{tag}l{i}.{proj}.w", cfg.hidden, cfg.hidden
            )
            sd[q + f"self_attn.{proj}.bias"] = _randn(f"{tag}l{i}.{proj}.b", cfg.hidden)
        sd[q + "layer_norm1.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm1.bias"] = torch.zeros(cfg.hidden)
        sd[q + "layer_norm2.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm2.bias"] = torch.zeros(cfg.hidden)
        sd[q + "mlp.fc1.weight"] = _randn(f"{tag}l{i}.fc1.w", cfg.intermediate, cfg.hidden)
        sd[q + "mlp.fc1.bias"] = _randn(f"{tag}l{i}.fc1.b", cfg.intermediate)
        sd[q + "mlp.fc2.weight"] = _randn(f"{tag}l{i}.fc2.w", cfg.hidden, cfg.intermediate)
        sd[q + "mlp.fc2.bias"] = _randn(f"{tag}l{i}.fc2.b", cfg.hidden)
    sd[p + "post_layernorm.weight"] = torch.ones(cfg.hidden)
    sd[p + "post_layernorm.bias"] = torch.zeros(cfg.hidden)
    sd["visual_projection.weight"] = _randn(tag + "vproj", cfg.proj, cfg.hidden)
    return sd


def make_clip_vit_b32_weights() -> dict[str, torch.Tensor]:
    """Flagship (ViT-B/32) weights — original entry point.

    NOTE: generated under the legacy un-tagged seed names so existing
    golden-pinned tests and the smoke cosine stay bit-identical.
    """
    cfg = VIT_B32
    sd: dict[str, torch.Tensor] = {}
    p = "vision_model."
    sd[p + "embeddings.class_embedding"] = _randn("cls", cfg.hidden)
    sd[p + "embeddings.patch_embedding.weight"] = _randn(
        "patch", cfg.hidden, 3, cfg.patch, cfg.patch
    )
    sd[p + "embeddings.position_embedding.weight"] = _randn("pos", cfg.num_pos, cfg.hidden)
    sd[p + "pre_layrnorm.weight"] = torch.ones(cfg.hidden)
    sd[p + "pre_layrnorm.bias"] = torch.zeros(cfg.hidden)
    for i in range(cfg.layers):
        q = f"{p}encoder.layers.{i}."
        for proj in ["q_proj", "k_proj", "v_proj", "out_proj"]:
            sd[q + f"self_attn.{proj}.weight"] = _randn(f"l{i}.{proj}.w", cfg.hidden, cfg.hidden)
            sd[q + f"self_attn.{proj}.bias"] = _randn(f"l{i}.{proj}.b", cfg.hidden)
        sd[q + "layer_norm1.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm1.bias"] = torch.zeros(cfg.hidden)
        sd[q + "layer_norm2.weight"] = torch.ones(cfg.hidden)
        sd[q + "layer_norm2.bias"] = torch.zeros(cfg.hidden)
        sd[q + "mlp.fc1.weight"] = _randn(f"l{i}.fc1.w", cfg.intermediate, cfg.hidden)
        sd[q + "mlp.fc1.bias"] = _randn(f"l{i}.fc1.b", cfg.intermediate)
        sd[q + "mlp.fc2.weight"] = _randn(f"l{i}.fc2.w", cfg.hidden, cfg.intermediate)
        sd[q + "mlp.fc2.bias"] = _randn(f"l{i}.fc2.b", cfg.hidden)
    sd[p + "post_layernorm.weight"] = torch.ones(cfg.hidden)
    sd[p + "post_layernorm.bias"] = torch.zeros(cfg.hidden)
    sd["visual_projection.weight"] = _randn("vproj", cfg.proj, cfg.hidden)
    return sd


def make_clip_vit_weights_stress(cfg: VitConfig = VIT_B32) -> dict[str, torch.Tensor]:
    """Pretrained-LIKE weights: realistic scales + outlier channels.

    Seeded-random weights at sigma=0.02 produce tame activation ranges;
    pretrained CLIP checkpoints carry outlier channels whose activations
    reach the tens-to-hundreds, stressing bf16 rounding and the
    exp/rcp-based fused activation epilogues (VERDICT r01 "What's weak"
    #2).  This state dict injects those features synthetically:

    - a handful of fc1/fc2/out-proj channels scaled 16-48x (the
      "massive activations" pattern of pretrained ViTs),
    - LayerNorm gains spread in [0.3, 3] with a few ~8x outlier gains
      and non-zero biases,
    - class/position embeddings at pretrained-like norms.

    Used by the GPU stress parity test (cosine >= 0.999 vs the fp32
    oracle on the SAME dict) — kernel-correctness evidence for real
    checkpoints without needing the network.
    """
    sd = make_clip_vit_weights(cfg)
    tag = cfg.name + ":stress:"

    def outlier_rows(name: str, t: torch.Tensor, n_out: int, scale: float) -> torch.Tensor:
        g = torch.Generator().manual_seed(_seed_for(tag + name))
        rows = torch.randperm(t.shape[0], generator=g)[:n_out]
        t = t.clone()
        t[rows] *= scale
        return t

    p = "vision_model."
    sd[p + "embeddings.class_embedding"] = (
        _randn(tag + "cls", cfg.hidden, std=0.6))
    sd[p + "embeddings.position_embedding.weight"] = _randn(
        tag + "pos", cfg.num_pos, cfg.hidden, std=0.15)
    for i in range(cfg.layers):
        q = f"{p}encoder.layers.{i}."
        sd[q + "mlp.fc1.weight"] = outlier_rows(
            f"fc1.{i}", sd[q + "mlp.fc1.weight"], 4, 16.0)
        sd[q + "mlp.fc2.weight"] = outlier_rows(
            f"fc2.{i}", sd[q + "mlp.fc2.weight"], 2, 24.0)
        sd[q + "self_attn.out_proj.weight"] = outlier_rows(
            f"out.{i}", sd[q + "self_attn.out_proj.weight"], 2, 8.0)
        for ln in ("layer_norm1", "layer_norm2"):
            w = 0.3 + 2.7 * torch.rand(
                cfg.hidden,
                generator=torch.Generator().manual_seed(
                    _seed_for(f"{tag}{ln}.w.{i}")))
            w = outlier_rows(f"{ln}.{i}", w.unsqueeze(1), 3, 8.0).squeeze(1)
            b = _randn(f"{tag}{ln}.b.{i}", cfg.hidden, std=0.5)
            sd[q + f"{ln}.weight"] = w
            sd[q + f"{ln}.bias"] = b
    return sd


def load_pretrained_state_dict(path: str) -> dict[str, torch.Tensor]:
    """Load a real CLIP/SigLIP vision checkpoint from a LOCAL directory
    or file (no network: reference models/clip.py:33 downloads
    openai/clip-vit-large-patch14; here the operator provides the files).

    Accepts a HF-style model directory (model.safetensors /
    pytorch_model.bin, possibly sharded) or a single safetensors/bin
    file.  Keys are filtered to the vision tower + projection and common
    prefixes are normalized, so the result feeds ClipVisionTowerAMD /
    SiglipVisionTowerAMD (and the fp32 oracle) verbatim.
    """
    import pathlib

    p = pathlib.Path(path)
    files: list[pathlib.Path]
    if p.is_dir():
        files = sorted(p.glob("*.safetensors")) or sorted(p.glob("pytorch_model*.bin"))
        if not files:
            msg = f"no safetensors/bin checkpoint files under {p}"
            raise FileNotFoundError(msg)
    elif p.is_file():
        files = [p]
    else:
        msg = f"checkpoint path {p} does not exist"
        raise FileNotFoundError(msg)

    sd: dict[str, torch.Tensor] = {}
    for f in files:
        if f.suffix == ".safetensors":
            from safetensors.torch import load_file

            sd.update(load_file(str(f)))
        else:
            sd.update(torch.load(f, map_location="cpu", weights_only=True))

    # normalize: CLIPModel checkpoints prefix the tower keys we want;
    # text-side keys are dropped
    out: dict[str, torch.Tensor] = {}
    for k, v in sd.items():
        for strip in ("clip.", "model."):
            if k.startswith(strip):
                k = k[len(strip):]
        if k.startswith(("vision_model.", "visual_projection.")):
            out[k] = v.float()
    if not out:
        msg = "checkpoint holds no vision_model.* / visual_projection.* keys"
        raise ValueError(msg)
    return out
