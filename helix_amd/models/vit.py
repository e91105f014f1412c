"""ViT image encoder for vision embeddings (the SigLIP2 role of the
reference's kodit vision path, kodit_init.go:50-70: images indexed into
the same vector space used for retrieval).

MI355X-native: patchify is an unfold + one MFMA GEMM (ops.gemm_bf16),
the transformer reuses the bge encoder block (non-causal flash
attention + layer_norm + fused-GELU GEMMs) — no new kernels needed,
which is the point: the encoder infra serves text AND vision.

Weights are random-init offline (synthetic parity like every model
here). A faithful HF SigLIP import is deliberately NOT shipped: SigLIP
is pre-LN while these blocks are post-LN (bge layout), so a name-map
alone would silently change semantics; deployments with a checkpoint
should read it via engine/weights.py `iter_safetensors` into a pre-LN
variant first.
"""
from __future__ import annotations

import io
from dataclasses import dataclass
from typing import List

import torch
import torch.nn as nn

from helix_amd import ops
from helix_amd.models.bert import BertConfig, BertLayer


@dataclass
class ViTConfig:
    name: str = "siglip-base"
    image_size: int = 224
    patch_size: int = 16
    hidden_size: int = 768
    intermediate_size: int = 3072
    num_layers: int = 12
    num_heads: int = 12
    layer_norm_eps: float = 1e-6
    pooling: str = "mean"            # siglip pools with attention/mean

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2

    @property
    def patch_dim(self) -> int:
        return 3 * self.patch_size * self.patch_size


VIT_PRESETS = {
    "siglip-base": ViTConfig(),
    "tiny-vit": ViTConfig(name="tiny-vit", image_size=32, patch_size=8,
                          hidden_size=128, intermediate_size=256,
                          num_layers=2, num_heads=2),
}


def preprocess_image(data: bytes, image_size: int) -> torch.Tensor:
    """bytes (png/jpeg/...) -> [3, S, S] float in [-1, 1] (SigLIP's
    normalization)."""
    from PIL import Image
    img = Image.open(io.BytesIO(data)).convert("RGB")
    img = img.resize((image_size, image_size), Image.BILINEAR)
    import numpy as np
    arr = torch.from_numpy(
        np.asarray(img, dtype="float32").copy()) / 127.5 - 1.0
    return arr.permute(2, 0, 1).contiguous()


class ViTEmbeddingModel(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.cfg = cfg
        # the encoder blocks are bge blocks with a ViT-shaped config
        bcfg = BertConfig(
            name=cfg.name, vocab_size=64, hidden_size=cfg.hidden_size,
            intermediate_size=cfg.intermediate_size,
            num_layers=cfg.num_layers, num_heads=cfg.num_heads,
            max_position=cfg.num_patches,
            layer_norm_eps=cfg.layer_norm_eps)
        self.patch_w = nn.Parameter(
            torch.empty(cfg.hidden_size, cfg.patch_dim))
        self.patch_b = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.pos_emb = nn.Embedding(cfg.num_patches, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [BertLayer(bcfg) for _ in range(cfg.num_layers)])
        self.post_ln_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_ln_b = nn.Parameter(torch.zeros(cfg.hidden_size))

    def _patchify(self, images: torch.Tensor) -> torch.Tensor:
        """[B, 3, S, S] -> [B*P, patch_dim] rows."""
        p = self.cfg.patch_size
        B = images.shape[0]
        # unfold H and W into patch grids, flatten per patch
        u = images.unfold(2, p, p).unfold(3, p, p)   # [B,3,gh,gw,p,p]
        u = u.permute(0, 2, 3, 1, 4, 5).reshape(
            B * self.cfg.num_patches, self.cfg.patch_dim)
        return u

    @torch.inference_mode()
    def forward(self, images: torch.Tensor) -> torch.Tensor:
        """images: [B, 3, S, S] float -> [B, H] L2-normalized."""
        B = images.shape[0]
        P = self.cfg.num_patches
        dtype = self.patch_w.dtype
        x = ops.gemm_bf16(self._patchify(images).to(dtype),
                          self.patch_w, self.patch_b)
        pos = torch.arange(P, device=images.device).repeat(B)
        x = x + self.pos_emb(pos).to(dtype)
        cu = torch.arange(0, (B + 1) * P, P, dtype=torch.int32,
                          device=images.device)
        for layer in self.layers:
            x = layer(x, cu, P)
        x = ops.layer_norm(x, self.post_ln_w, self.post_ln_b,
                           self.cfg.layer_norm_eps)
        emb = x.view(B, P, -1).float().mean(1) \
            if self.cfg.pooling == "mean" else x.view(B, P, -1)[:, 0].float()
        return torch.nn.functional.normalize(emb, dim=-1)

    @torch.inference_mode()
    def embed_images(self, blobs: List[bytes]) -> torch.Tensor:
        device = next(self.parameters()).device
        imgs = torch.stack([
            preprocess_image(b, self.cfg.image_size) for b in blobs
        ]).to(device)
        return self.forward(imgs)

    @torch.inference_mode()
    def init_random(self, seed: int = 0):
        on_gpu = next(self.parameters()).is_cuda
        if on_gpu:
            torch.cuda.manual_seed(seed)
        g = None if on_gpu else torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if name.endswith(("_b", "ln_w", "ln_b")):
                continue
            if on_gpu:
                p.data.normal_(0.0, 0.02)
            else:
                t = torch.empty(p.shape, dtype=torch.float32)
                t.normal_(0.0, 0.02, generator=g)
                p.data.copy_(t.to(p.dtype))
        return self

    def memory_bytes(self) -> int:
        return sum(p.numel() * p.element_size()
                   for p in self.parameters())
