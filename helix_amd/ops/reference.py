"""Pure-PyTorch fp32 reference implementations of every HIP kernel.

These define the numerics contract: GPU tests compare the CDNA4 kernels
against these (run in fp32) within bf16 tolerance. They also serve the
CPU-only paths (unit tests, engine logic tests on this no-GPU container).
"""
from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * w.float()
    return out.to(x.dtype)


def fused_add_rms_norm(x: torch.Tensor, residual: torch.Tensor,
                       w: torch.Tensor, eps: float):
    """Returns (normed, new_residual); mirrors the in-place HIP op."""
    new_res = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(new_res, w, eps), new_res


def layer_norm(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
               eps: float) -> torch.Tensor:
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), w.float(), b.float(), eps).to(x.dtype)


def rotary_embedding(positions: torch.Tensor, q: torch.Tensor,
                     k: torch.Tensor, cos_sin: torch.Tensor, head_dim: int):
    """Neox-style rotate-half RoPE, applied out-of-place; returns (q, k)."""
    half = head_dim // 2
    cs = cos_sin[positions]  # [T, D]
    cos = cs[:, :half].float()  # [T, half]
    sin = cs[:, half:].float()

    def rot(t: torch.Tensor) -> torch.Tensor:
        T = t.shape[0]
        th = t.float().view(T, -1, head_dim)
        x1, x2 = th[..., :half], th[..., half:]
        c = cos.unsqueeze(1)
        s = sin.unsqueeze(1)
        out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
        return out.view(t.shape).to(t.dtype)

    return rot(q), rot(k)


def make_cos_sin_cache(head_dim: int, max_pos: int, base: float = 10000.0,
                       device="cpu",
                       rope_scaling: dict | None = None) -> torch.Tensor:
    """[max_pos, D] fp32 rows = [cos(0..D/2), sin(0..D/2)].

    rope_scaling supports the llama3.1 scheme (HF config.json
    rope_scaling: rope_type "llama3" with factor / low_freq_factor /
    high_freq_factor / original_max_position_embeddings): wavelengths
    longer than original/low_freq are divided by `factor`, shorter than
    original/high_freq are untouched, with a smooth ramp between —
    the long-context extension Llama-3.1 ships with."""
    import math
    half = head_dim // 2
    inv_freq = 1.0 / (base ** (torch.arange(half, dtype=torch.float32,
                                            device=device) / half))
    if rope_scaling and rope_scaling.get("rope_type", rope_scaling.get(
            "type", "")) == "llama3":
        factor = float(rope_scaling.get("factor", 8.0))
        lo = float(rope_scaling.get("low_freq_factor", 1.0))
        hi = float(rope_scaling.get("high_freq_factor", 4.0))
        orig = float(rope_scaling.get(
            "original_max_position_embeddings", 8192))
        wavelen = 2 * math.pi / inv_freq
        low_wl = orig / lo
        high_wl = orig / hi
        scaled = inv_freq / factor
        # smooth interpolation in the medium-frequency band
        smooth = (orig / wavelen - lo) / (hi - lo)
        smooth = smooth.clamp(0.0, 1.0)
        mid = (1 - smooth) * scaled + smooth * inv_freq
        inv_freq = torch.where(wavelen > low_wl, scaled,
                               torch.where(wavelen < high_wl,
                                           inv_freq, mid))
    t = torch.arange(max_pos, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)  # [max_pos, half]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    i = x.shape[-1] // 2
    g, u = x[..., :i].float(), x[..., i:].float()
    return (torch.nn.functional.silu(g) * u).to(x.dtype)


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x.float(), approximate="tanh").to(x.dtype)


def reshape_and_cache(k: torch.Tensor, v: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slot_mapping: torch.Tensor):
    """k/v: [T, Hkv, D]; caches: [nblocks, Hkv, bs, D]."""
    bs = k_cache.shape[2]
    for t in range(slot_mapping.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        b, off = slot // bs, slot % bs
        k_cache[b, :, off] = k[t]
        v_cache[b, :, off] = v[t]


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 cu_seqlens: torch.Tensor, max_seqlen: int,
                 scale: float, causal: bool = True,
                 window: int = 0, cu_seqlens_k=None) -> torch.Tensor:
    """Varlen causal GQA attention (optional sliding window and
    cached-prefix query offset). q: [T, Hq, D], k/v: [Tk, Hkv, D]."""
    Hq, Hkv = q.shape[1], k.shape[1]
    G = Hq // Hkv
    if cu_seqlens_k is None:
        cu_seqlens_k = cu_seqlens
    out = torch.empty_like(q)
    for i in range(cu_seqlens.shape[0] - 1):
        qs0, qe = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        ks0, ke = int(cu_seqlens_k[i]), int(cu_seqlens_k[i + 1])
        qlen, klen = qe - qs0, ke - ks0
        ctx = klen - qlen
        qs = q[qs0:qe].float()
        ks = k[ks0:ke].float().repeat_interleave(G, dim=1)
        vs = v[ks0:ke].float().repeat_interleave(G, dim=1)
        qpos = torch.arange(ctx, klen)
        kpos = torch.arange(klen)
        if causal:
            m = kpos[None, :] <= qpos[:, None]
            if window > 0:
                m &= (qpos[:, None] - kpos[None, :]) < window
        else:
            m = torch.ones(qlen, klen, dtype=torch.bool)
        mask = torch.where(m, 0.0, float("-inf"))
        o = torch.nn.functional.scaled_dot_product_attention(
            qs.transpose(0, 1), ks.transpose(0, 1), vs.transpose(0, 1),
            attn_mask=mask, scale=scale)
        out[qs0:qe] = o.transpose(0, 1).to(q.dtype)
    return out


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float,
                      window: int = 0) -> torch.Tensor:
    """q: [B, Hq, D] single token per seq. Gathers KV then full attention."""
    B, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    bs = k_cache.shape[2]
    G = Hq // Hkv
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        nb = (L + bs - 1) // bs
        blocks = block_tables[b, :nb].long()
        k = k_cache[blocks].permute(0, 2, 1, 3).reshape(nb * bs, Hkv, D)[:L]
        v = v_cache[blocks].permute(0, 2, 1, 3).reshape(nb * bs, Hkv, D)[:L]
        if window > 0:
            k = k[max(0, L - window):]
            v = v[max(0, L - window):]
        kf = k.float().repeat_interleave(G, dim=1)  # [L, Hq, D]
        vf = v.float().repeat_interleave(G, dim=1)
        qf = q[b].float()  # [Hq, D]
        s = torch.einsum("hd,lhd->hl", qf, kf) * scale
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hl,lhd->hd", p, vf)
        out[b] = o.to(q.dtype)
    return out


def sample_tokens(logits: torch.Tensor, temperatures: torch.Tensor,
                  seeds: torch.Tensor) -> torch.Tensor:
    """Greedy for T<=0; Gumbel-max for T>0 driven by the per-row seeds.

    Seed-determinism matters even on CPU: SPMD-TP ranks each run this
    sampler and must pick identical tokens (the HIP kernel derives its
    Gumbel noise from the same seeds; distribution-equal, not bit-equal
    — GPU tests check greedy exactly and stochastic statistically).
    """
    out = torch.empty(logits.shape[0], dtype=torch.int64,
                      device=logits.device)
    for i in range(logits.shape[0]):
        t = float(temperatures[i])
        row = logits[i].float()
        if t <= 0:
            out[i] = int(row.argmax())
        else:
            g = torch.Generator()
            g.manual_seed(int(seeds[i]) & 0x7FFFFFFFFFFFFFFF)
            u = torch.rand(row.shape[0], generator=g)
            gumbel = -torch.log(-torch.log(u.clamp(min=1e-20)))
            out[i] = int((row / t + gumbel).argmax())
    return out


def gemm_bf16(x: torch.Tensor, w: torch.Tensor, bias=None,
              act: int = 0) -> torch.Tensor:
    out = x.float() @ w.float().t()
    if bias is not None:
        out = out + bias.float()
    if act == 1:
        out = torch.nn.functional.gelu(out, approximate="tanh")
    return out.to(x.dtype)
