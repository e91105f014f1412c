"""Llama-family decoder (Llama-3 8B/70B, Mistral-7B) on helix_amd CDNA4 ops.

Replaces the model execution the reference delegates to vLLM containers
(SURVEY.md §2.8): hand-written HIP kernels for RMSNorm/RoPE/attention/
activation; plain projection GEMMs via hipBLASLt (torch.linear); paged KV.

Tensor-parallel sharding (RCCL over xGMI) is applied by
helix_amd.parallel when tp_size > 1: column-parallel QKV/gate_up,
row-parallel o_proj/down_proj with all-reduce.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from helix_amd import ops
from helix_amd.ops import make_cos_sin_cache


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rms_norm_eps: float = 1e-5
    rope_base: float = 500000.0
    max_position: int = 8192
    tie_embeddings: bool = False
    sliding_window: int = 0        # 0 = full attention (Mistral v0.1: 4096)
    attention_bias: bool = False   # Qwen2-style QKV bias
    rope_scaling: dict = field(default_factory=dict)  # llama3.1 scheme

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


PRESETS = {
    "llama3-8b": LlamaConfig(),
    # Llama-3.1: same weights shape as 3.0 but 128k context via the
    # llama3 rope-scaling scheme (HF config.json rope_scaling)
    "llama3.1-8b": LlamaConfig(
        name="llama3.1-8b", max_position=32768, rope_base=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192}),
    "llama3-70b": LlamaConfig(
        name="llama3-70b", hidden_size=8192, intermediate_size=28672,
        num_layers=80, num_heads=64, num_kv_heads=8),
    "mistral-7b": LlamaConfig(
        name="mistral-7b", vocab_size=32000, rope_base=10000.0,
        sliding_window=4096),
    # Tiny configs for CPU tests and smoke runs.
    "tiny": LlamaConfig(
        name="tiny", vocab_size=512, hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
        max_position=512, rope_base=10000.0),
    "tiny-gqa": LlamaConfig(
        name="tiny-gqa", vocab_size=1024, hidden_size=512,
        intermediate_size=1024, num_layers=4, num_heads=8, num_kv_heads=2,
        head_dim=64, max_position=1024, rope_base=10000.0),
    "qwen2-7b": LlamaConfig(
        name="qwen2-7b", vocab_size=152064, hidden_size=3584,
        intermediate_size=18944, num_layers=28, num_heads=28,
        num_kv_heads=4, head_dim=128, rope_base=1000000.0,
        max_position=32768, attention_bias=True),
    "tiny-qwen": LlamaConfig(
        name="tiny-qwen", vocab_size=512, hidden_size=256,
        intermediate_size=512, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=64, max_position=512, rope_base=10000.0,
        attention_bias=True),
    "tiny-sw": LlamaConfig(
        name="tiny-sw", vocab_size=512, hidden_size=256,
        intermediate_size=512, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=64, max_position=512, rope_base=10000.0,
        sliding_window=24),
}


@dataclass
class PrefillMeta:
    """Varlen prefill: q/k/v are concatenated new tokens of all sequences.
    With prefix caching, q covers only the uncached suffix; attention
    gathers the full context K/V from the paged cache (gather_blk/off)
    and cu_seqlens_k carries the full-context boundaries."""
    cu_seqlens: torch.Tensor       # [B+1] int32 (new-token boundaries)
    max_seqlen: int
    slot_mapping: torch.Tensor     # [T] int64 cache slots for new tokens
    positions: torch.Tensor        # [T] int64
    cu_seqlens_k: Optional[torch.Tensor] = None   # [B+1] int32 full context
    gather_blk: Optional[torch.Tensor] = None     # [Tk] int64
    gather_off: Optional[torch.Tensor] = None     # [Tk] int64
    is_prefill: bool = True


@dataclass
class DecodeMeta:
    """One new token per running sequence."""
    block_tables: torch.Tensor     # [B, max_blocks] int32
    seq_lens: torch.Tensor         # [B] int32 (context incl. current token)
    slot_mapping: torch.Tensor     # [B] int64
    positions: torch.Tensor        # [B] int64
    max_len: int = 0               # host-side max(seq_lens) (avoids sync)
    workspace: Optional[tuple] = None
    is_prefill: bool = False


class HLinear(nn.Linear):
    """nn.Linear routed through ops.linear: decode-shaped (skinny-M)
    GEMMs run the owned split-K MFMA kernel, prefill-shaped ones stay on
    hipBLASLt (SURVEY §2.8 'Q/K/V + O projections, MLP GEMMs')."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.linear(x, self.weight, self.bias)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_size: int = 1, tp_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        assert cfg.num_heads % tp_size == 0 and cfg.num_kv_heads % tp_size == 0
        self.nh = cfg.num_heads // tp_size
        self.nkv = cfg.num_kv_heads // tp_size
        self.hd = cfg.head_dim
        self.scale = self.hd ** -0.5
        self.tp_size = tp_size
        self.window = cfg.sliding_window or 0
        q, kv, h = self.nh * self.hd, self.nkv * self.hd, cfg.hidden_size
        self.qkv_proj = HLinear(h, q + 2 * kv, bias=cfg.attention_bias)
        self.o_proj = HLinear(q, h, bias=False)

    def forward(self, x: torch.Tensor, cos_sin: torch.Tensor, kv_cache,
                meta) -> torch.Tensor:
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        # fused strided split + rope + paged-cache write (one kernel, no
        # .contiguous() copies); compact K/V emitted only when the
        # fresh-prefill attention path consumes them directly
        want_kv = meta.is_prefill and \
            getattr(meta, "cu_seqlens_k", None) is None
        q, k, v = ops.rope_qkv_cache(
            meta.positions, qkv, cos_sin, self.nh, self.nkv, self.hd,
            kv_cache=kv_cache, slot_mapping=meta.slot_mapping,
            want_kv=want_kv)
        if meta.is_prefill:
            if meta.cu_seqlens_k is not None:
                # cached-prefix prefill: full-context K/V gathered from
                # the paged cache (new tokens were just written above)
                k_cache, v_cache = kv_cache
                k_all = k_cache[meta.gather_blk, :, meta.gather_off]
                v_all = v_cache[meta.gather_blk, :, meta.gather_off]
                if k_all.dtype == torch.uint8:      # fp8 KV cache
                    k_all = ops.kv_fp8_dequant(k_all, q.dtype)
                    v_all = ops.kv_fp8_dequant(v_all, q.dtype)
                o = ops.attn_prefill(q, k_all, v_all, meta.cu_seqlens,
                                     meta.max_seqlen, self.scale,
                                     window=self.window,
                                     cu_seqlens_k=meta.cu_seqlens_k)
            else:
                o = ops.attn_prefill(q, k, v, meta.cu_seqlens,
                                     meta.max_seqlen, self.scale,
                                     window=self.window)
        else:
            k_cache, v_cache = kv_cache
            o = ops.paged_attn_decode(q, k_cache, v_cache, meta.block_tables,
                                      meta.seq_lens, self.scale,
                                      meta.workspace,
                                      meta.max_len or None,
                                      window=self.window)
        out = self.o_proj(o.view(T, -1))
        if self.tp_size > 1:
            from helix_amd import parallel
            out = parallel.tp_all_reduce(out)
        return out


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_size: int = 1):
        super().__init__()
        assert cfg.intermediate_size % tp_size == 0
        self.tp_size = tp_size
        i = cfg.intermediate_size // tp_size
        self.gate_up_proj = HLinear(cfg.hidden_size, 2 * i, bias=False)
        self.down_proj = HLinear(i, cfg.hidden_size, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.down_proj(ops.silu_and_mul(self.gate_up_proj(x)))
        if self.tp_size > 1:
            from helix_amd import parallel
            out = parallel.tp_all_reduce(out)
        return out


class DecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_size: int = 1, tp_rank: int = 0):
        super().__init__()
        self.input_norm_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_norm_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.attn = Attention(cfg, tp_size, tp_rank)
        self.mlp = MLP(cfg, tp_size)
        self.eps = cfg.rms_norm_eps

    def forward(self, x, residual, cos_sin, kv_cache, meta):
        if residual is None:
            residual = x.clone()
            h = ops.rms_norm(x, self.input_norm_w, self.eps)
        else:
            h, residual = ops.fused_add_rms_norm(x, residual,
                                                 self.input_norm_w, self.eps)
        h = self.attn(h, cos_sin, kv_cache, meta)
        h, residual = ops.fused_add_rms_norm(h, residual, self.post_norm_w,
                                             self.eps)
        h = self.mlp(h)
        return h, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_size: int = 1, tp_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp_size = tp_size
        self.tp_rank = tp_rank
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [DecoderLayer(cfg, tp_size, tp_rank) for _ in range(cfg.num_layers)])
        self.final_norm_w = nn.Parameter(torch.ones(cfg.hidden_size))
        if cfg.tie_embeddings:
            self.lm_head = None
        else:
            self.lm_head = HLinear(cfg.hidden_size, cfg.vocab_size,
                                   bias=False)
        cs = make_cos_sin_cache(cfg.head_dim, cfg.max_position, cfg.rope_base,
                                rope_scaling=cfg.rope_scaling or None)
        self.register_buffer("cos_sin", cs, persistent=False)

    def _apply(self, fn, recurse=True):
        # keep the RoPE table fp32 through .to(bfloat16) casts — the HIP
        # kernel consumes fp32 cos/sin.
        super()._apply(fn, recurse)
        if self.cos_sin.dtype != torch.float32:
            self.cos_sin = self.cos_sin.float()
        return self

    @torch.inference_mode()
    def forward(self, input_ids: torch.Tensor, kv_caches, meta) -> torch.Tensor:
        x = self.embed_tokens(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            x, residual = layer(x, residual, self.cos_sin, cache, meta)
        x = (x.float() + residual.float()).to(x.dtype)
        return ops.rms_norm(x, self.final_norm_w, self.cfg.rms_norm_eps)

    @torch.inference_mode()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.lm_head is not None:
            return self.lm_head(hidden)
        return F.linear(hidden, self.embed_tokens.weight)

    @torch.inference_mode()
    def init_random(self, seed: int = 0):
        """Random-init weights (no network => no real checkpoints).

        CPU path is generator-deterministic (tests build identical twins);
        GPU path inits in-place at device speed (8B+ scale).
        """
        std = 0.02
        on_gpu = next(self.parameters()).is_cuda
        if on_gpu:
            torch.cuda.manual_seed(seed)
        g = None if on_gpu else torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if name.endswith("norm_w"):
                p.data.fill_(1.0)
            elif name.endswith(".bias"):
                p.data.zero_() if on_gpu else p.data.copy_(
                    torch.zeros(p.shape, dtype=torch.float32).to(p.dtype))
            elif on_gpu:
                p.data.normal_(0.0, std)
            else:
                t = torch.empty(p.shape, dtype=torch.float32)
                t.normal_(0.0, std, generator=g)
                p.data.copy_(t.to(p.dtype))
        return self

    def memory_bytes(self) -> int:
        return sum(p.numel() * p.element_size() for p in self.parameters())
