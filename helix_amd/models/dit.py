"""Diffusion image generation — closes the last delegated op in
SURVEY.md §2.8: the reference hands `/v1/images/generations` to a
diffusers container (inferenceproxy/proxy.go:113; runtime enum
types/runner.go:86). Here it is a native engine.

MI355X-first rectified-flow DiT ("flux-lite" architecture class):

- tiny conv VAE (8x spatial down, 4 latent channels) — decode is a few
  small convs, negligible next to the transformer;
- in-context conditioning: prompt tokens (shared tokenizer + learned
  embedding) are prepended to the latent-patch token stream, so one
  non-causal varlen flash-attention kernel serves text AND image tokens
  — the same `ops.attn_prefill` the bge/ViT encoders run;
- DiT blocks with adaLN modulation from (timestep, pooled prompt);
  every projection runs on the hand-written MFMA bf16 GEMM
  (`ops.gemm_bf16`), norms on `ops.layer_norm`;
- Euler integration of the rectified-flow ODE (velocity prediction,
  sigma-shifted schedule), deterministic per seed: noise is drawn on a
  CPU generator so CPU and GPU runs of the same seed share a prior.

Weights are random-init offline like every model family here (synthetic
parity; no network for checkpoints) — the full serving path (tokenize ->
condition -> iterate ODE -> VAE decode -> PNG) is real.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Callable, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from helix_amd import ops


@dataclass
class DiTConfig:
    name: str = "flux-lite"
    image_size: int = 256
    vae_down: int = 8                # spatial downsampling of the VAE
    latent_ch: int = 4
    vae_ch: int = 64                 # base conv width
    patch: int = 2                   # DiT patch size on the latent grid
    hidden: int = 1024
    depth: int = 12
    heads: int = 16                  # head_dim 64 = the flash kernel tile
    text_len: int = 32
    vocab_size: int = 66304          # padded /64 like the LLM vocab
    time_dim: int = 256              # sinusoidal timestep embedding width
    eps: float = 1e-6

    @property
    def latent_size(self) -> int:
        return self.image_size // self.vae_down

    @property
    def num_patches(self) -> int:
        return (self.latent_size // self.patch) ** 2

    @property
    def patch_dim(self) -> int:
        return self.latent_ch * self.patch * self.patch

    @property
    def seq_len(self) -> int:
        return self.text_len + self.num_patches


DIT_PRESETS = {
    "flux-lite": DiTConfig(),
    "tiny-dit": DiTConfig(name="tiny-dit", image_size=32, vae_ch=16,
                          hidden=128, depth=2, heads=2, text_len=8,
                          vocab_size=512, time_dim=64),
}


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    """Sinusoidal embedding of t in [0, 1] (scaled x1000, DiT's
    convention); returns [B, dim] float32."""
    half = dim // 2
    freqs = torch.exp(
        -math.log(10000.0) *
        torch.arange(half, dtype=torch.float32, device=t.device) / half)
    args = t.float()[:, None] * 1000.0 * freqs[None]
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1)


class TinyVAE(nn.Module):
    """Conv autoencoder: [B,3,S,S] <-> [B,latent_ch,S/8,S/8]. Decode is
    the hot direction (one call per generated image)."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        c = cfg.vae_ch
        self.enc = nn.Sequential(
            nn.Conv2d(3, c, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(c, 2 * c, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(2 * c, 2 * c, 3, stride=2, padding=1), nn.SiLU(),
            nn.Conv2d(2 * c, cfg.latent_ch, 3, padding=1))
        self.dec = nn.Sequential(
            nn.Conv2d(cfg.latent_ch, 2 * c, 3, padding=1), nn.SiLU(),
            nn.Upsample(scale_factor=2, mode="nearest"),
            nn.Conv2d(2 * c, 2 * c, 3, padding=1), nn.SiLU(),
            nn.Upsample(scale_factor=2, mode="nearest"),
            nn.Conv2d(2 * c, c, 3, padding=1), nn.SiLU(),
            nn.Upsample(scale_factor=2, mode="nearest"),
            nn.Conv2d(c, 3, 3, padding=1))

    def encode(self, images: torch.Tensor) -> torch.Tensor:
        # NHWC keeps MIOpen on its igemm bf16 conv path from the first
        # call (NCHW bf16 runs a naive kernel until MIOpen's find
        # completes; e2e steady state measured identical — see
        # profiles/r02_imagegen.md postscript)
        return self.enc(images.contiguous(
            memory_format=torch.channels_last))

    def decode(self, latents: torch.Tensor) -> torch.Tensor:
        return self.dec(latents.contiguous(
            memory_format=torch.channels_last)).contiguous()


class DiTBlock(nn.Module):
    """Pre-norm transformer block with adaLN modulation (shift/scale/
    gate x attn/mlp from the conditioning vector)."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        h = cfg.hidden
        self.cfg = cfg
        self.qkv_w = nn.Parameter(torch.empty(3 * h, h))
        self.qkv_b = nn.Parameter(torch.zeros(3 * h))
        self.out_w = nn.Parameter(torch.empty(h, h))
        self.out_b = nn.Parameter(torch.zeros(h))
        self.up_w = nn.Parameter(torch.empty(4 * h, h))
        self.up_b = nn.Parameter(torch.zeros(4 * h))
        self.down_w = nn.Parameter(torch.empty(h, 4 * h))
        self.down_b = nn.Parameter(torch.zeros(h))
        self.mod_w = nn.Parameter(torch.empty(6 * h, h))
        self.mod_b = nn.Parameter(torch.zeros(6 * h))
        # affine-free LayerNorm operands for ops.layer_norm
        self.register_buffer("ln_w", torch.ones(h), persistent=False)
        self.register_buffer("ln_b", torch.zeros(h), persistent=False)
        self.hd = h // cfg.heads
        self.scale = self.hd ** -0.5

    def forward(self, x: torch.Tensor, cond: torch.Tensor,
                cu: torch.Tensor, L: int) -> torch.Tensor:
        h = self.cfg.hidden
        T = x.shape[0]
        mod = ops.gemm_bf16(cond, self.mod_w, self.mod_b)
        sa, ca, ga, sm, cm, gm = (m.repeat_interleave(L, dim=0)
                                  for m in mod.chunk(6, dim=-1))
        n = ops.layer_norm(x, self.ln_w, self.ln_b, self.cfg.eps)
        n = (n * (1 + ca) + sa).to(x.dtype)
        qkv = ops.gemm_bf16(n, self.qkv_w, self.qkv_b)
        q, k, v = qkv.split([h, h, h], dim=-1)
        nh = self.cfg.heads
        o = ops.attn_prefill(q.contiguous().view(T, nh, self.hd),
                             k.contiguous().view(T, nh, self.hd),
                             v.contiguous().view(T, nh, self.hd),
                             cu, L, self.scale, causal=False)
        a = ops.gemm_bf16(o.view(T, h), self.out_w, self.out_b)
        x = (x.float() + ga.float() * a.float()).to(x.dtype)
        n = ops.layer_norm(x, self.ln_w, self.ln_b, self.cfg.eps)
        n = (n * (1 + cm) + sm).to(x.dtype)
        m = ops.gemm_bf16(n, self.up_w, self.up_b, act=1)     # fused GELU
        m = ops.gemm_bf16(m, self.down_w, self.down_b)
        return (x.float() + gm.float() * m.float()).to(x.dtype)


class DiTModel(nn.Module):
    """Velocity field v(x_t, t, prompt) on the latent grid."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        self.cfg = cfg
        h = cfg.hidden
        # patch rows are zero-padded to a K multiple of 64 — the native
        # MFMA GEMM's K-granularity (gemm_bf16.hip); padded weight
        # columns multiply zeros and are inert
        self.in_k = ((cfg.patch_dim + 63) // 64) * 64
        self.in_w = nn.Parameter(torch.empty(h, self.in_k))
        self.in_b = nn.Parameter(torch.zeros(h))
        self.txt_emb = nn.Embedding(cfg.vocab_size, h)
        self.pos_emb = nn.Embedding(cfg.seq_len, h)
        self.t_w1 = nn.Parameter(torch.empty(h, cfg.time_dim))
        self.t_b1 = nn.Parameter(torch.zeros(h))
        self.t_w2 = nn.Parameter(torch.empty(h, h))
        self.t_b2 = nn.Parameter(torch.zeros(h))
        self.blocks = nn.ModuleList(
            [DiTBlock(cfg) for _ in range(cfg.depth)])
        self.fin_mod_w = nn.Parameter(torch.empty(2 * h, h))
        self.fin_mod_b = nn.Parameter(torch.zeros(2 * h))
        self.fin_w = nn.Parameter(torch.empty(cfg.patch_dim, h))
        self.fin_b = nn.Parameter(torch.zeros(cfg.patch_dim))
        self.register_buffer("ln_w", torch.ones(h), persistent=False)
        self.register_buffer("ln_b", torch.zeros(h), persistent=False)

    def _patchify(self, lat: torch.Tensor) -> torch.Tensor:
        p = self.cfg.patch
        B = lat.shape[0]
        u = lat.unfold(2, p, p).unfold(3, p, p)      # [B,C,gh,gw,p,p]
        u = u.permute(0, 2, 3, 1, 4, 5).reshape(
            B * self.cfg.num_patches, self.cfg.patch_dim)
        return u

    def _unpatchify(self, rows: torch.Tensor, B: int) -> torch.Tensor:
        cfg = self.cfg
        p, s = cfg.patch, cfg.latent_size
        g = s // p
        u = rows.view(B, g, g, cfg.latent_ch, p, p)
        return u.permute(0, 3, 1, 4, 2, 5).reshape(B, cfg.latent_ch, s, s)

    def forward(self, lat: torch.Tensor, t: torch.Tensor,
                text_ids: torch.Tensor) -> torch.Tensor:
        """lat [B,C,s,s], t [B] in [0,1], text_ids [B,text_len] ->
        velocity [B,C,s,s] float32."""
        cfg = self.cfg
        B = lat.shape[0]
        P, Tt, L = cfg.num_patches, cfg.text_len, cfg.seq_len
        dtype = self.in_w.dtype
        rows = self._patchify(lat).to(dtype)
        if self.in_k != cfg.patch_dim:
            rows = F.pad(rows, (0, self.in_k - cfg.patch_dim))
        x_img = ops.gemm_bf16(rows.contiguous(), self.in_w, self.in_b)
        tx = self.txt_emb(text_ids.clamp(0, cfg.vocab_size - 1))
        x = torch.cat([tx, x_img.view(B, P, -1)], dim=1).view(B * L, -1)
        pos = torch.arange(L, device=lat.device).repeat(B)
        x = (x + self.pos_emb(pos)).to(dtype)
        temb = timestep_embedding(t, cfg.time_dim).to(dtype)
        cond = ops.gemm_bf16(temb, self.t_w1, self.t_b1, act=1)
        cond = ops.gemm_bf16(cond, self.t_w2, self.t_b2)
        cond = (cond.float() + tx.float().mean(dim=1)).to(dtype)
        cu = torch.arange(0, (B + 1) * L, L, dtype=torch.int32,
                          device=lat.device)
        for blk in self.blocks:
            x = blk(x, cond, cu, L)
        n = ops.layer_norm(x, self.ln_w, self.ln_b, cfg.eps)
        fm = ops.gemm_bf16(cond, self.fin_mod_w, self.fin_mod_b)
        shift, scale = (m.repeat_interleave(L, dim=0)
                        for m in fm.chunk(2, dim=-1))
        n = (n * (1 + scale) + shift).to(dtype)
        out = ops.gemm_bf16(n, self.fin_w, self.fin_b)
        patch_rows = out.view(B, L, -1)[:, Tt:, :].reshape(
            B * P, cfg.patch_dim)
        return self._unpatchify(patch_rows.float(), B)


def rf_schedule(steps: int, shift: float = 3.0) -> torch.Tensor:
    """Sigma-shifted rectified-flow timestep schedule 1 -> 0 (flux's
    resolution-dependent shift, fixed here per preset)."""
    ts = torch.linspace(1.0, 0.0, steps + 1)
    return shift * ts / (1 + (shift - 1) * ts)


@torch.inference_mode()
def rf_sample(v_fn: Callable[[torch.Tensor, float], torch.Tensor],
              shape, steps: int, seed: int,
              device, shift: float = 3.0) -> torch.Tensor:
    """Euler-integrate dx/dt = v(x, t) from t=1 (noise) to t=0 (data).
    With x_t = (1-t)x0 + t*n and v = n - x0 the integrator is exact for
    an oracle velocity regardless of step count (tested). Noise comes
    from a CPU generator so a seed means the same image on CPU and GPU.
    """
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(shape, generator=g).to(device=device,
                                           dtype=torch.float32)
    ts = rf_schedule(steps, shift)
    for i in range(steps):
        t, tn = float(ts[i]), float(ts[i + 1])
        v = v_fn(x, t)
        x = x + (tn - t) * v.float()
    return x


class DiffusionImageModel(nn.Module):
    """Full text->image pipeline: DiT velocity field + VAE decode."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        self.cfg = cfg
        self.dit = DiTModel(cfg)
        self.vae = TinyVAE(cfg).to(memory_format=torch.channels_last)

    @torch.inference_mode()
    def generate(self, prompts_ids: List[List[int]], steps: int = 8,
                 seed: int = 0) -> torch.Tensor:
        """Token-id prompts -> [B, 3, S, S] uint8 images."""
        cfg = self.cfg
        B = len(prompts_ids)
        device = next(self.parameters()).device
        ids = torch.zeros(B, cfg.text_len, dtype=torch.int64,
                          device=device)
        for i, p in enumerate(prompts_ids):
            p = p[:cfg.text_len]
            if p:
                ids[i, :len(p)] = torch.tensor(
                    p, dtype=torch.int64, device=device)
        tvec = torch.empty(B, device=device)

        def v_fn(x, t):
            tvec.fill_(t)
            return self.dit(x.to(next(self.parameters()).dtype),
                            tvec, ids)

        lat = rf_sample(v_fn, (B, cfg.latent_ch, cfg.latent_size,
                               cfg.latent_size), steps, seed, device)
        img = self.vae.decode(lat.to(next(self.parameters()).dtype))
        img = (img.float().clamp(-1, 1) + 1.0) * 127.5
        return img.round().to(torch.uint8)

    @torch.inference_mode()
    def init_random(self, seed: int = 0):
        on_gpu = next(self.parameters()).is_cuda
        if on_gpu:
            torch.cuda.manual_seed(seed)
        g = None if on_gpu else torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if name.endswith(("_b", "ln_w", "ln_b", ".bias")):
                continue
            # conv weights get variance-preserving init so the VAE
            # stack neither saturates nor flattens random-init images
            # (seed/prompt sensitivity is part of the synthetic parity)
            std = (p.numel() / p.shape[0]) ** -0.5 if p.dim() == 4 \
                else 0.02
            if on_gpu:
                p.data.normal_(0.0, std)
            else:
                t = torch.empty(p.shape, dtype=torch.float32)
                t.normal_(0.0, std, generator=g)
                p.data.copy_(t.to(p.dtype))
        return self

    def memory_bytes(self) -> int:
        return sum(p.numel() * p.element_size()
                   for p in self.parameters())
