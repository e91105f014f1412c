"""bge-class BERT encoder for embeddings (feeds the RAG vector store).

Replaces the reference's delegated embedding models (kodit ONNX text
model / `/v1/embeddings` proxying, SURVEY.md §2.8 "Embedding models").
All projection GEMMs run on the hand-written MFMA bf16 GEMM
(ops.gemm_bf16, gemm_bf16.hip) with fused bias + GELU on the MLP up
projection; attention is the non-causal mode of the varlen flash kernel.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from helix_amd import ops


@dataclass
class BertConfig:
    name: str = "bge-base"
    vocab_size: int = 30528          # padded to /64 for the MFMA GEMM
    hidden_size: int = 768
    intermediate_size: int = 3072
    num_layers: int = 12
    num_heads: int = 12
    max_position: int = 512
    layer_norm_eps: float = 1e-12
    pooling: str = "cls"             # cls | mean


BERT_PRESETS = {
    "bge-base": BertConfig(),
    "bge-large": BertConfig(name="bge-large", hidden_size=1024,
                            intermediate_size=4096, num_layers=24,
                            num_heads=16),
    "tiny-bert": BertConfig(name="tiny-bert", vocab_size=512, hidden_size=128,
                            intermediate_size=256, num_layers=2, num_heads=2,
                            max_position=128),
}


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        h = cfg.hidden_size
        self.qkv_w = nn.Parameter(torch.empty(3 * h, h))
        self.qkv_b = nn.Parameter(torch.zeros(3 * h))
        self.attn_out_w = nn.Parameter(torch.empty(h, h))
        self.attn_out_b = nn.Parameter(torch.zeros(h))
        self.attn_ln_w = nn.Parameter(torch.ones(h))
        self.attn_ln_b = nn.Parameter(torch.zeros(h))
        self.up_w = nn.Parameter(torch.empty(cfg.intermediate_size, h))
        self.up_b = nn.Parameter(torch.zeros(cfg.intermediate_size))
        self.down_w = nn.Parameter(torch.empty(h, cfg.intermediate_size))
        self.down_b = nn.Parameter(torch.zeros(h))
        self.out_ln_w = nn.Parameter(torch.ones(h))
        self.out_ln_b = nn.Parameter(torch.zeros(h))
        self.cfg = cfg
        self.hd = cfg.hidden_size // cfg.num_heads
        self.scale = self.hd ** -0.5

    def forward(self, x, cu, max_len):
        T = x.shape[0]
        h = self.cfg.hidden_size
        qkv = ops.gemm_bf16(x, self.qkv_w, self.qkv_b)
        q, k, v = qkv.split([h, h, h], dim=-1)
        nh = self.cfg.num_heads
        o = ops.attn_prefill(q.contiguous().view(T, nh, self.hd),
                             k.contiguous().view(T, nh, self.hd),
                             v.contiguous().view(T, nh, self.hd),
                             cu, max_len, self.scale, causal=False)
        a = ops.gemm_bf16(o.view(T, h), self.attn_out_w, self.attn_out_b)
        x = ops.layer_norm((x.float() + a.float()).to(x.dtype),
                           self.attn_ln_w, self.attn_ln_b,
                           self.cfg.layer_norm_eps)
        m = ops.gemm_bf16(x, self.up_w, self.up_b, act=1)   # fused GELU
        m = ops.gemm_bf16(m, self.down_w, self.down_b)
        return ops.layer_norm((x.float() + m.float()).to(x.dtype),
                              self.out_ln_w, self.out_ln_b,
                              self.cfg.layer_norm_eps)


class BertEmbeddingModel(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.word_emb = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.pos_emb = nn.Embedding(cfg.max_position, cfg.hidden_size)
        self.emb_ln_w = nn.Parameter(torch.ones(cfg.hidden_size))
        self.emb_ln_b = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.layers = nn.ModuleList(
            [BertLayer(cfg) for _ in range(cfg.num_layers)])

    @torch.inference_mode()
    def forward(self, input_ids: torch.Tensor, cu_seqlens: torch.Tensor,
                max_len: int) -> torch.Tensor:
        """input_ids: [T] varlen concat; returns [B, H] L2-normalized."""
        positions = torch.cat([
            torch.arange(int(cu_seqlens[i + 1] - cu_seqlens[i]),
                         device=input_ids.device)
            for i in range(cu_seqlens.shape[0] - 1)])
        x = self.word_emb(input_ids) + self.pos_emb(positions)
        x = ops.layer_norm(x, self.emb_ln_w, self.emb_ln_b,
                           self.cfg.layer_norm_eps)
        for layer in self.layers:
            x = layer(x, cu_seqlens, max_len)
        outs = []
        for i in range(cu_seqlens.shape[0] - 1):
            s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
            if self.cfg.pooling == "mean":
                outs.append(x[s:e].float().mean(0))
            else:
                outs.append(x[s].float())
        emb = torch.stack(outs)
        return torch.nn.functional.normalize(emb, dim=-1)

    @torch.inference_mode()
    def init_random(self, seed: int = 0):
        on_gpu = next(self.parameters()).is_cuda
        if on_gpu:
            torch.cuda.manual_seed(seed)
        g = None if on_gpu else torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if name.endswith(("_b", "ln_w", "ln_b")):
                continue  # keep zeros/ones defaults
            if on_gpu:
                p.data.normal_(0.0, 0.02)
            else:
                t = torch.empty(p.shape, dtype=torch.float32)
                t.normal_(0.0, 0.02, generator=g)
                p.data.copy_(t.to(p.dtype))
        return self

    def memory_bytes(self) -> int:
        return sum(p.numel() * p.element_size() for p in self.parameters())
