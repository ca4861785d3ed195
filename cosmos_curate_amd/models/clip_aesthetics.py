"""Aesthetic scorer — CLIP embeddings -> tiny MLP score.

Mirror of /root/reference/cosmos_curate/models/clip_aesthetics.py:27-80
(``CLIPAestheticScorer`` chaining CLIPImageEmbeddings + AestheticScorer)
and models/aesthetics.py:30-80 (the sac-logos-ava1 MLP
in_dim->1024->128->64->16->1 with dropout placeholders, inference-only).

The rebuild's CLIP tower emits 512-d (ViT-B/32) rather than the
reference's 768-d (ViT-L/14), so the MLP's first layer is 512 wide; the
published MLP weights are unavailable offline, so weights are name-seeded
(same stand-in convention as clip_weights.py).  The CLIP ViT MFMA kernels
are shared — the filter adds only a 512x1024 GEMM per frame batch
(SURVEY.md §2 "Aesthetic filter": reuses ViT kernels; MLP trivial).
"""

from __future__ import annotations

import hashlib

import numpy as np
import numpy.typing as npt
import torch
from torch import nn

from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.models.clip import CLIPImageEmbeddings

_AESTHETICS_MODEL_ID = "ttj/sac-logos-ava1-l14-linearMSE"
_CLIP_MODEL_ID = "openai/clip-vit-base-patch32"


class MLP(nn.Module):
    """aesthetics.py:30-80 head (512-d input on the rebuild)."""

    def __init__(self, in_dim: int = 512) -> None:
        super().__init__()
        self.layers = nn.Sequential(
            nn.Linear(in_dim, 1024),
            nn.Dropout(0.2),
            nn.Linear(1024, 128),
            nn.Dropout(0.2),
            nn.Linear(128, 64),
            nn.Dropout(0.1),
            nn.Linear(64, 16),
            nn.Linear(16, 1),
        )

    @torch.no_grad()
    def forward(self, embed: torch.Tensor) -> torch.Tensor:
        return self.layers(embed)


def _seeded(name: str, *shape: int) -> torch.Tensor:
    seed = int.from_bytes(hashlib.sha256(("aes:" + name).encode()).digest()[:8], "little")
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g) * 0.05


def make_aesthetic_weights(in_dim: int = 512) -> dict[str, torch.Tensor]:
    mlp = MLP(in_dim)
    return {
        name: (_seeded(name, *t.shape) if t.ndim > 1 else torch.zeros_like(t))
        for name, t in mlp.state_dict().items()
    }


class AestheticScorer(ModelInterface):
    """aesthetics.py:84-140 surface."""

    def __init__(self) -> None:
        super().__init__()
        self._mlp: MLP | None = None

    @property
    def conda_env_name(self) -> str:
        return "unified"

    @property
    def model_id_names(self) -> list[str]:
        return [_AESTHETICS_MODEL_ID]

    def setup(self) -> None:
        self._mlp = MLP()
        self._mlp.load_state_dict(make_aesthetic_weights())
        self._mlp.eval()
        if torch.cuda.is_available():
            self._mlp.cuda()

    @torch.no_grad()
    def __call__(self, embeddings: torch.Tensor) -> torch.Tensor:
        assert self._mlp is not None, "setup() not called"
        return self._mlp(embeddings.float()).squeeze(-1)


class CLIPAestheticScorer(ModelInterface):
    """clip_aesthetics.py:27-80: frames -> CLIP embeds -> score per frame."""

    def __init__(self) -> None:
        super().__init__()
        self._clip_model: CLIPImageEmbeddings | None = None
        self._aesthetic_model: AestheticScorer | None = None

    @property
    def conda_env_name(self) -> str:
        return "unified"

    @property
    def model_id_names(self) -> list[str]:
        return [_AESTHETICS_MODEL_ID, _CLIP_MODEL_ID]

    def setup(self) -> None:
        self._clip_model = CLIPImageEmbeddings()
        self._aesthetic_model = AestheticScorer()
        self._clip_model.setup()
        self._aesthetic_model.setup()

    def __call__(self, images: torch.Tensor | npt.NDArray[np.uint8]) -> torch.Tensor:
        assert self._clip_model and self._aesthetic_model
        return self._aesthetic_model(self._clip_model(images))
