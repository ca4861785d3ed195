"""Pretrained-checkpoint readiness (CPU): the loader round-trips real
checkpoint formats and the state-dict keys are exactly what transformers
(the reference's own arithmetic, models/clip.py:33) expects."""

from __future__ import annotations

import pytest
import torch

from cosmos_curate_amd.models import clip_weights as cw


def test_loader_roundtrip_safetensors(tmp_path):
    from safetensors.torch import save_file

    sd = cw.make_clip_vit_weights_stress(cw.VIT_B32)
    save_file({k: v.contiguous() for k, v in sd.items()},
              str(tmp_path / "model.safetensors"))
    got = cw.load_pretrained_state_dict(str(tmp_path))
    assert set(got) == set(sd)
    for k in sd:
        torch.testing.assert_close(got[k], sd[k])


def test_loader_strips_clipmodel_prefix_and_drops_text(tmp_path):
    from safetensors.torch import save_file

    sd = cw.make_clip_vit_weights(cw.VIT_B32)
    mixed = {f"clip.{k}": v.contiguous() for k, v in sd.items()}
    mixed["text_model.embeddings.token_embedding.weight"] = torch.zeros(4, 4)
    save_file(mixed, str(tmp_path / "model.safetensors"))
    got = cw.load_pretrained_state_dict(str(tmp_path))
    assert set(got) == set(sd)


def test_loader_missing_path_fails_loudly(tmp_path):
    with pytest.raises(FileNotFoundError):
        cw.load_pretrained_state_dict(str(tmp_path / "nope"))
    (tmp_path / "empty").mkdir()
    with pytest.raises(FileNotFoundError):
        cw.load_pretrained_state_dict(str(tmp_path / "empty"))


@pytest.mark.parametrize("cfg", [cw.VIT_B32, cw.VIT_L14],
                         ids=lambda c: c.name)
def test_keys_match_transformers_exactly(cfg):
    """load_state_dict(strict=True) into the real transformers module:
    zero missing, zero unexpected — a pretrained checkpoint of this
    geometry loads verbatim on both the oracle and the product tower."""
    transformers = pytest.importorskip("transformers")

    config = transformers.CLIPVisionConfig(
        hidden_size=cfg.hidden,
        intermediate_size=cfg.intermediate,
        num_hidden_layers=cfg.layers,
        num_attention_heads=cfg.heads,
        image_size=cfg.image,
        patch_size=cfg.patch,
        projection_dim=cfg.proj,
    )
    model = transformers.CLIPVisionModelWithProjection(config)
    sd = (cw.make_clip_vit_b32_weights() if cfg is cw.VIT_B32
          else cw.make_clip_vit_weights(cfg))
    result = model.load_state_dict(sd, strict=True)
    assert not result.missing_keys and not result.unexpected_keys
