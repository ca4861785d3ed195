"""CLIPImageEmbeddings ModelInterface — the rebuild's flagship embedder.

Mirror of /root/reference/cosmos_curate/models/clip.py:36-118: same class
name, same ModelInterface surface (conda_env_name / model_id_names /
setup / __call__ over u8 NHWC frames or NCHW tensors), same output contract
(L2-normalized image embeddings, clip.py:71-74).  Inside: the CLIP
preprocess chain runs as one fused HIP kernel (cc_clip_preprocess) and the
ViT forward on the MFMA GEMM path (clip_vit.ClipVisionTowerAMD), bf16.

At the benchmark config frames arrive already resized to 224x224
(clip_frame_extraction target_res, SURVEY.md §2 row "Clip frame
extraction"), so torchvision's Resize/CenterCrop (clip.py:48-56) are
identity; inputs of other sizes are resized on device with the bicubic
kernel first (torchvision BICUBIC semantics differ from cv2 INTER_CUBIC
only in antialias, which is OFF for upscaling and irrelevant at 224->224).
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt
import torch

from cosmos_curate_amd import hotpath
from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights

_CLIP_MODEL_ID = "openai/clip-vit-base-patch32"

CLIP_MEAN = (0.48145466, 0.4578275, 0.40821073)
CLIP_STD = (0.26862954, 0.26130258, 0.27577711)
SIGLIP_MEAN = (0.5, 0.5, 0.5)  # SiglipImageProcessor defaults
SIGLIP_STD = (0.5, 0.5, 0.5)


class _CLIPImageEmbeddings(torch.nn.Module):
    """Device-side implementation (reference clip.py:36-74 counterpart)."""

    def __init__(self, variant: str = "vit_b32",
                 checkpoint_path: str | None = None) -> None:
        super().__init__()
        import os

        from cosmos_curate_amd.models import clip_weights as cw

        hotpath.require_gpu()  # fail loudly before any lazy surprises
        self.device = torch.device("cuda")
        cfg = cw.CONFIGS[variant]
        # real checkpoints (reference models/clip.py:33 downloads them;
        # here the operator provides local files — no network)
        checkpoint_path = checkpoint_path or os.environ.get("CC_CLIP_CHECKPOINT")
        pretrained = (cw.load_pretrained_state_dict(checkpoint_path)
                      if checkpoint_path else None)
        if variant.startswith("siglip"):
            from cosmos_curate_amd.models.siglip_vit import SiglipVisionTowerAMD

            self.tower = SiglipVisionTowerAMD(
                pretrained or cw.make_siglip_weights(cfg), cfg).to(self.device)
            self._mean_arr = np.array(SIGLIP_MEAN, dtype=np.float32)
            self._std_arr = np.array(SIGLIP_STD, dtype=np.float32)
        else:
            sd = pretrained or (
                make_clip_vit_b32_weights() if variant == "vit_b32"
                else cw.make_clip_vit_weights(cfg))
            self.tower = ClipVisionTowerAMD(sd, cfg).to(self.device)
            self._mean_arr = np.array(CLIP_MEAN, dtype=np.float32)
            self._std_arr = np.array(CLIP_STD, dtype=np.float32)

    def preprocess_u8(self, frames_dev_u8: torch.Tensor) -> torch.Tensor:
        """(N,224,224,3) u8 on device -> (N,3,224,224) bf16 normalized."""
        import ctypes

        lib = hotpath.require_gpu()
        n, h, w, _ = frames_dev_u8.shape
        out = torch.empty((n, 3, h, w), dtype=torch.bfloat16, device=frames_dev_u8.device)
        stream = torch.cuda.current_stream(frames_dev_u8.device).cuda_stream
        mean = (ctypes.c_float * 3)(*self._mean_arr)
        std = (ctypes.c_float * 3)(*self._std_arr)
        hotpath.check(
            lib.cc_clip_preprocess(
                frames_dev_u8.contiguous().data_ptr(), n, h, w, mean, std,
                out.data_ptr(), 1, stream,
            )
        )
        return out

    def preprocess_patches_u8(self, frames_dev_u8: torch.Tensor) -> torch.Tensor:
        """(N,224,224,3) u8 on device -> GEMM-ready patch rows
        [N*g*g, patch_k] bf16 in one fused kernel (normalize +
        patch-extract; replaces preprocess_u8 + the tower's torch
        reshape/permute/pad)."""
        import ctypes

        lib = hotpath.require_gpu()
        n, h, w, _ = frames_dev_u8.shape
        cfg = self.tower.cfg
        g = cfg.image // cfg.patch
        out = torch.empty(
            (n * g * g, self.tower.patch_k), dtype=torch.bfloat16,
            device=frames_dev_u8.device,
        )
        stream = torch.cuda.current_stream(frames_dev_u8.device).cuda_stream
        mean = (ctypes.c_float * 3)(*self._mean_arr)
        std = (ctypes.c_float * 3)(*self._std_arr)
        hotpath.check(
            lib.cc_clip_preprocess_patches(
                frames_dev_u8.contiguous().data_ptr(), n, h, w, cfg.patch,
                self.tower.patch_k, mean, std, out.data_ptr(), stream,
            )
        )
        return out

    def _resize_center_crop_u8(self, frames_dev_u8: torch.Tensor) -> torch.Tensor:
        """(N,H,W,3) u8 device -> (N,image,image,3) u8: bicubic resize of
        the shorter side to cfg.image + center crop (torchvision
        Resize(BICUBIC)+CenterCrop chain, reference clip.py:48-56), on
        device via cc_resize_bicubic_u8."""
        import ctypes  # noqa: F401 — hotpath ctypes setup already done

        lib = hotpath.require_gpu()
        n, h, w, _ = frames_dev_u8.shape
        size = self.tower.cfg.image
        scale = size / min(h, w)
        rh, rw = max(size, round(h * scale)), max(size, round(w * scale))
        out = torch.empty((n, rh, rw, 3), dtype=torch.uint8,
                          device=frames_dev_u8.device)
        stream = torch.cuda.current_stream(frames_dev_u8.device).cuda_stream
        hotpath.check(
            lib.cc_resize_bicubic_u8(
                frames_dev_u8.contiguous().data_ptr(), n, h, w,
                out.data_ptr(), rh, rw, stream,
            )
        )
        top = (rh - size) // 2
        left = (rw - size) // 2
        return out[:, top:top + size, left:left + size, :].contiguous()

    @torch.no_grad()
    def __call__(self, images: torch.Tensor | npt.NDArray[np.uint8]) -> torch.Tensor:
        if isinstance(images, np.ndarray):
            # (N,H,W,C) u8 host array (reference clip.py:66-68 entry form)
            images = torch.from_numpy(np.ascontiguousarray(images)).to(self.device)
        if (
            images.dtype == torch.uint8 and images.ndim == 4
            and images.shape[-1] == 3
        ):
            images = images.to(self.device)
            if (images.shape[1] != self.tower.cfg.image
                    or images.shape[2] != self.tower.cfg.image):
                # target_res=-1 runs hand source-resolution frames here;
                # resize+crop on device first (docstring contract above)
                images = self._resize_center_crop_u8(images)
            n = images.shape[0]
            patches = self.preprocess_patches_u8(images)
            return self.tower(patches=patches, n=n)
        pixels = images.to(self.device, dtype=torch.bfloat16)
        return self.tower(pixels)


class CLIPImageEmbeddings(ModelInterface):
    """ModelInterface wrapper (reference clip.py:77-118).

    variant "vit_b32" (flagship) | "vit_l14" (the reference's own CLIP
    model id, openai/clip-vit-large-patch14 — BASELINE config #3 class).
    """

    def __init__(self, variant: str = "vit_b32",
                 checkpoint_path: str | None = None) -> None:
        super().__init__()
        self._variant = variant
        self._checkpoint_path = checkpoint_path
        self._model: _CLIPImageEmbeddings | None = None

    @property
    def conda_env_name(self) -> str:
        return "unified"  # constant on the rebuild (single ROCm env)

    @property
    def model_id_names(self) -> list[str]:
        if self._variant == "vit_l14":
            return ["openai/clip-vit-large-patch14"]  # reference clip.py:33
        if self._variant.startswith("siglip"):
            return ["google/siglip-large-patch16-256"]  # BASELINE config #3
        return [_CLIP_MODEL_ID]

    def setup(self) -> None:
        self._model = _CLIPImageEmbeddings(self._variant,
                                           checkpoint_path=self._checkpoint_path)

    def __call__(self, images: torch.Tensor | npt.NDArray[np.uint8]) -> torch.Tensor:
        assert self._model is not None, "setup() not called"
        return self._model(images)
