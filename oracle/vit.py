"""CLIP ViT embedding oracle (torch-cpu fp32).

The reference embeds frames with HuggingFace ``CLIPModel.get_image_features``
plus L2 normalization (/root/reference/cosmos_curate/models/clip.py:64-74).
transformers is installed in this image, so the oracle runs the *reference's
own arithmetic* (transformers CLIP vision tower, fp32, CPU) -- only the
weights differ: HF hub weights are unavailable offline, so both oracle and
product use the same fixed-seed random-init weights produced by
``cosmos_curate_amd.models.clip_weights`` (BASELINE.md measurement plan).

Parity contract (BASELINE.json): product bf16 MFMA path within >=0.999
cosine of this fp32 oracle on identical inputs+weights.
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt
import torch


def build_reference_clip_vision(state_dict: dict[str, torch.Tensor]):
    """Build a transformers CLIP vision tower + projection from a state dict.

    The state dict uses transformers' own key names
    (``vision_model.*`` / ``visual_projection.weight``), as produced by
    ``cosmos_curate_amd.models.clip_weights.make_clip_vit_b32_weights``.
    """
    from transformers import CLIPVisionConfig
    from transformers.models.clip.modeling_clip import CLIPVisionModelWithProjection

    cfg = CLIPVisionConfig()  # defaults = ViT-B/32: hidden 768, patch 32, 12 layers
    model = CLIPVisionModelWithProjection(cfg)
    missing, unexpected = model.load_state_dict(state_dict, strict=False)
    missing = [m for m in missing if "position_ids" not in m]
    assert not missing and not unexpected, (missing, unexpected)
    return model.float().eval()


@torch.no_grad()
def embed_frames_fp32(
    model, pixel_values: npt.NDArray[np.float32] | torch.Tensor
) -> npt.NDArray[np.float32]:
    """(N,3,224,224) f32 CLIP-normalized -> (N,512) L2-normalized fp32 embeds.

    Mirrors models/clip.py:64-74: get_image_features then
    embed / ||embed||.
    """
    if isinstance(pixel_values, np.ndarray):
        pixel_values = torch.from_numpy(pixel_values)
    out = model(pixel_values=pixel_values.float()).image_embeds
    out = out / torch.linalg.vector_norm(out, dim=-1, keepdim=True)
    return out.cpu().numpy().astype(np.float32)


def build_reference_siglip_vision(state_dict: dict[str, torch.Tensor],
                                  cfg_overrides: dict | None = None):
    """Build a transformers SiglipVisionModel from a state dict made by
    clip_weights.make_siglip_weights (key names are exactly
    SiglipVisionModel.named_parameters()).  Default geometry =
    google/siglip-large-patch16-256."""
    from transformers import SiglipVisionConfig, SiglipVisionModel

    kw = dict(hidden_size=1024, intermediate_size=4096, num_hidden_layers=24,
              num_attention_heads=16, image_size=256, patch_size=16)
    if cfg_overrides:
        kw.update(cfg_overrides)
    model = SiglipVisionModel(SiglipVisionConfig(**kw))
    missing, unexpected = model.load_state_dict(state_dict, strict=False)
    missing = [m for m in missing if "position_ids" not in m]
    assert not missing and not unexpected, (missing, unexpected)
    return model.float().eval()


@torch.no_grad()
def siglip_embed_frames_fp32(
    model, pixel_values: npt.NDArray[np.float32] | torch.Tensor
) -> npt.NDArray[np.float32]:
    """(N,3,image,image) f32 SigLIP-normalized -> L2-normalized
    MAP-pooled fp32 embeddings (pooler_output / ||.||)."""
    if isinstance(pixel_values, np.ndarray):
        pixel_values = torch.from_numpy(pixel_values)
    out = model(pixel_values=pixel_values.float()).pooler_output
    out = out / torch.linalg.vector_norm(out, dim=-1, keepdim=True)
    return out.cpu().numpy().astype(np.float32)
