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
