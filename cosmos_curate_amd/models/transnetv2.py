"""TransNetV2 shot-transition detector — MI355X rebuild.

From-scratch torch implementation of the architecture the reference uses
(/root/reference/cosmos_curate/models/transnetv2.py:39-528, itself after
Soucek & Lokoc 2020, arXiv:2008.04838): 3 stacked DDCNN blocks of 4-way
dilated (2+1)D convolutions with dense shortcuts + frame-similarity and
color-histogram heads over a 101-frame lookup window -> per-frame
transition probability.  Module names mirror the reference so state dicts
interchange; parity vs the reference implementation is pinned by committed
golden vectors (tests/golden/transnetv2_golden.npz, generated in the dev
container by oracle/gen_transnet_golden.py).

The network is tiny ((1,100,27,48,3) input) and runs on torch-rocm/MIOpen
(not kernel-worthy).  Published weights
(Sn4kehead/TransNetV2) are unavailable offline; the stand-in is the
name-seeded deterministic generator below (same convention as
clip_weights.py).
"""

from __future__ import annotations

import hashlib

import torch
from torch import nn
from torch.nn import functional as F


class Conv3DConfigurable(nn.Module):
    """(2+1)D separable conv pair (reference transnetv2.py:279-347)."""

    def __init__(self, in_filters: int, filters: int, dilation_rate: int,
                 *, use_bias: bool = True) -> None:
        super().__init__()
        spatial = nn.Conv3d(in_filters, 2 * filters, kernel_size=(1, 3, 3),
                            padding=(0, 1, 1), bias=False)
        temporal = nn.Conv3d(2 * filters, filters, kernel_size=(3, 1, 1),
                             dilation=(dilation_rate, 1, 1),
                             padding=(dilation_rate, 0, 0), bias=use_bias)
        self.layers = nn.ModuleList([spatial, temporal])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for layer in self.layers:
            x = layer(x)
        return x


class DilatedDCNNV2(nn.Module):
    """4 dilation rates (1/2/4/8) concatenated + BN (reference :224-277)."""

    def __init__(self, in_filters: int, filters: int, *,
                 activation: bool = True) -> None:
        super().__init__()
        self.Conv3D_1 = Conv3DConfigurable(in_filters, filters, 1, use_bias=False)
        self.Conv3D_2 = Conv3DConfigurable(in_filters, filters, 2, use_bias=False)
        self.Conv3D_4 = Conv3DConfigurable(in_filters, filters, 4, use_bias=False)
        self.Conv3D_8 = Conv3DConfigurable(in_filters, filters, 8, use_bias=False)
        self.bn = nn.BatchNorm3d(filters * 4, eps=1e-3)
        self._activation = activation

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = torch.cat(
            [self.Conv3D_1(x), self.Conv3D_2(x), self.Conv3D_4(x), self.Conv3D_8(x)],
            dim=1,
        )
        y = self.bn(y)
        return F.relu(y) if self._activation else y


class StackedDDCNNV2(nn.Module):
    """2 DDCNN blocks + dense shortcut + 2x2 avg pool (reference :152-222)."""

    def __init__(self, in_filters: int, n_blocks: int, filters: int) -> None:
        super().__init__()
        self.DDCNN = nn.ModuleList(
            [
                DilatedDCNNV2(
                    in_filters if i == 0 else filters * 4, filters,
                    activation=(i != n_blocks - 1),
                )
                for i in range(n_blocks)
            ]
        )
        self.pool = nn.AvgPool3d(kernel_size=(1, 2, 2))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shortcut = None
        for block in self.DDCNN:
            x = block(x)
            if shortcut is None:
                shortcut = x
        x = F.relu(x)
        x = x + shortcut
        return self.pool(x)


def _window_gather(sim: torch.Tensor, lookup: int) -> torch.Tensor:
    """(B,T,T) similarity -> (B,T,lookup) windows centered per frame."""
    half = (lookup - 1) // 2
    padded = F.pad(sim, [half, half])
    b, t = sim.shape[0], sim.shape[1]
    ti = torch.arange(t, device=sim.device).view(1, t, 1)
    li = torch.arange(lookup, device=sim.device).view(1, 1, lookup) + ti
    bi = torch.arange(b, device=sim.device).view(b, 1, 1).expand(b, t, lookup)
    return padded[bi, ti.expand(b, t, lookup), li.expand(b, t, lookup)]


class FrameSimilarity(nn.Module):
    """Projected cosine-similarity window head (reference :349-420)."""

    def __init__(self, in_filters: int, similarity_dim: int = 128,
                 lookup_window: int = 101, output_dim: int = 128,
                 *, use_bias: bool = True) -> None:
        super().__init__()
        self.projection = nn.Linear(in_filters, similarity_dim, bias=use_bias)
        self.fc = nn.Linear(lookup_window, output_dim)
        self.lookup_window = lookup_window

    def forward(self, block_features: list[torch.Tensor]) -> torch.Tensor:
        x = torch.cat([f.mean(dim=[3, 4]) for f in block_features], dim=1)
        x = x.transpose(1, 2)
        x = F.normalize(self.projection(x), p=2, dim=2)
        sim = torch.bmm(x, x.transpose(1, 2))
        return F.relu(self.fc(_window_gather(sim, self.lookup_window)))


class ColorHistograms(nn.Module):
    """512-bin RGB histogram similarity head (reference :422-528)."""

    def __init__(self, lookup_window: int = 101, output_dim: int = 128) -> None:
        super().__init__()
        self.fc = nn.Linear(lookup_window, output_dim)
        self.lookup_window = lookup_window

    @staticmethod
    def _histograms(frames_u8: torch.Tensor) -> torch.Tensor:
        f = frames_u8.int()
        b, t, h, w, _ = f.shape
        flat = f.view(b * t, h * w, 3)
        bins = ((flat[:, :, 0] >> 5) << 6) + ((flat[:, :, 1] >> 5) << 3) + (flat[:, :, 2] >> 5)
        prefix = (torch.arange(b * t, device=f.device) << 9).view(-1, 1)
        idx = (bins + prefix).view(-1)
        hist = torch.zeros(b * t * 512, dtype=torch.int32, device=f.device)
        hist.scatter_add_(0, idx, torch.ones_like(idx, dtype=torch.int32))
        return F.normalize(hist.view(b, t, 512).float(), p=2, dim=2)

    def forward(self, frames_u8: torch.Tensor) -> torch.Tensor:
        x = self._histograms(frames_u8)
        sim = torch.bmm(x, x.transpose(1, 2))
        return F.relu(self.fc(_window_gather(sim, self.lookup_window)))


class _TransNetV2(nn.Module):
    """Full detector (reference :39-149): u8 (B,T,27,48,3) -> (B,T,1) prob."""

    def __init__(self, rf: int = 16, rl: int = 3, rs: int = 2, rd: int = 1024) -> None:
        super().__init__()
        self.SDDCNN = nn.ModuleList(
            [StackedDDCNNV2(3, rs, rf)]
            + [StackedDDCNNV2(rf * 2 ** (i - 1) * 4, rs, rf * 2**i) for i in range(1, rl)]
        )
        self.frame_sim_layer = FrameSimilarity(
            sum(rf * 2**i * 4 for i in range(rl)), 128, 101, 128, use_bias=True
        )
        self.color_hist_layer = ColorHistograms(101, 128)
        out_dim = rf * 2 ** (rl - 1) * 4 * 3 * 6 + 128 + 128
        self.fc1 = nn.Linear(out_dim, rd)
        self.cls_layer1 = nn.Linear(rd, 1)
        self.cls_layer2 = nn.Linear(rd, 1)  # many-hot aux head (unused at inference)
        self.eval()

    @torch.no_grad()
    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        assert inputs.dtype == torch.uint8 and tuple(inputs.shape[2:]) == (27, 48, 3)
        x = inputs.permute(0, 4, 1, 2, 3).float() / 255.0
        feats = []
        for block in self.SDDCNN:
            x = block(x)
            feats.append(x)
        x = x.permute(0, 2, 3, 4, 1).reshape(x.shape[0], x.shape[2], -1)
        x = torch.cat([self.frame_sim_layer(feats), x], dim=2)
        x = torch.cat([self.color_hist_layer(inputs), x], dim=2)
        x = F.relu(self.fc1(x))
        return torch.sigmoid(self.cls_layer1(x))


def _seed_for(name: str) -> int:
    return int.from_bytes(hashlib.sha256(("tnv2:" + name).encode()).digest()[:8], "little")


def make_transnetv2_weights() -> dict[str, torch.Tensor]:
    """Deterministic name-seeded stand-in weights (offline; see module doc)."""
    net = _TransNetV2()
    sd = {}
    for name, t in net.state_dict().items():
        g = torch.Generator().manual_seed(_seed_for(name))
        if name.endswith("num_batches_tracked"):
            sd[name] = torch.zeros_like(t)
        elif "bn.running_var" in name:
            sd[name] = (torch.randn(t.shape, generator=g).abs() * 0.1 + 0.9)
        elif "bn.running_mean" in name:
            sd[name] = torch.randn(t.shape, generator=g) * 0.05
        elif name.endswith(".bias"):
            sd[name] = torch.randn(t.shape, generator=g) * 0.02
        else:
            sd[name] = torch.randn(t.shape, generator=g) * 0.05
    return sd


from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface  # noqa: E402

_TRANSNETV2_MODEL_ID = "Sn4kehead/TransNetV2"


class TransNetV2(ModelInterface):
    """ModelInterface wrapper (reference transnetv2.py:530-600 surface)."""

    def __init__(self) -> None:
        super().__init__()
        self._model: _TransNetV2 | None = None

    @property
    def conda_env_name(self) -> str:
        return "unified"

    @property
    def model_id_names(self) -> list[str]:
        return [_TRANSNETV2_MODEL_ID]

    def setup(self) -> None:
        self._model = _TransNetV2()
        self._model.load_state_dict(make_transnetv2_weights())
        self._model.eval()
        if torch.cuda.is_available():
            self._model.cuda()

    def __call__(self, inputs: torch.Tensor) -> torch.Tensor:
        assert self._model is not None, "setup() not called"
        with torch.no_grad():
            return self._model(inputs)
