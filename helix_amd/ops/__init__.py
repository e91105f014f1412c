"""helix_amd.ops — CDNA4 (gfx950) kernel dispatch.

GPU tensors run the hand-written HIP kernels in helix_amd._C and FAIL
LOUDLY if the extension is missing (no silent eager fallback on a GPU
box). CPU tensors use the pure-torch reference implementations so engine
logic is testable without a GPU.
"""
from __future__ import annotations

import os

import torch

from . import reference as ref
from .reference import make_cos_sin_cache  # re-export

_C = None
_C_ERR: str | None = None
try:
    from helix_amd import _C  # type: ignore  # noqa: F401
    from helix_amd import _C as _C_mod
    _C = _C_mod
except ImportError as e:  # pragma: no cover
    _C_ERR = str(e)


def _native():
    if _C is None:
        raise RuntimeError(
            "helix_amd._C native extension is not built but a GPU tensor was "
            "passed. Build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_C_ERR}")
    return _C


def have_native() -> bool:
    return _C is not None


def rms_norm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().rms_norm(out, x, w, eps)
        return out
    return ref.rms_norm(x, w, eps)


def fused_add_rms_norm(x: torch.Tensor, residual: torch.Tensor,
                       w: torch.Tensor, eps: float):
    """In-place on GPU: x <- norm(x+res), residual <- x+res. Returns both."""
    if x.is_cuda:
        _native().fused_add_rms_norm(x, residual, w, eps)
        return x, residual
    out, new_res = ref.fused_add_rms_norm(x, residual, w, eps)
    x.copy_(out)
    residual.copy_(new_res)
    return x, residual


def layer_norm(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
               eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().layer_norm(out, x, w, b, eps)
        return out
    return ref.layer_norm(x, w, b, eps)


def rotary_embedding(positions: torch.Tensor, q: torch.Tensor,
                     k: torch.Tensor, cos_sin: torch.Tensor, head_dim: int):
    """In-place on GPU. Returns (q, k)."""
    if q.is_cuda:
        _native().rotary_embedding(positions, q, k, cos_sin, head_dim)
        return q, k
    qo, ko = ref.rotary_embedding(positions, q, k, cos_sin, head_dim)
    q.copy_(qo)
    k.copy_(ko)
    return q, k


def kv_fp8_quant(t: torch.Tensor) -> torch.Tensor:
    """bf16 -> OCP e4m3 bytes (uint8 view), any shape."""
    return t.to(torch.float8_e4m3fn).view(torch.uint8)


def kv_fp8_dequant(t: torch.Tensor, dtype=torch.bfloat16) -> torch.Tensor:
    """OCP e4m3 bytes (uint8) -> dtype."""
    return t.view(torch.float8_e4m3fn).to(dtype)


def rope_qkv_cache(positions: torch.Tensor, qkv: torch.Tensor,
                   cos_sin: torch.Tensor, num_q: int, num_kv: int,
                   head_dim: int, kv_cache=None, slot_mapping=None,
                   want_kv: bool = False):
    """Fused QKV epilogue: strided split + rope + paged-cache write.

    Returns (q [T, num_q, head_dim], k, v) — k/v are None unless
    want_kv (fresh-prefill path needs compact K/V for attn_prefill).
    """
    T = qkv.shape[0]
    if qkv.is_cuda:
        q_out = torch.empty(T, num_q * head_dim, dtype=qkv.dtype,
                            device=qkv.device)
        k_out = v_out = None
        if want_kv:
            k_out = torch.empty(T, num_kv * head_dim, dtype=qkv.dtype,
                                device=qkv.device)
            v_out = torch.empty_like(k_out)
        kc = vc = None
        if kv_cache is not None:
            kc, vc = kv_cache
        _native().rope_qkv_cache(positions, qkv, q_out, kc, vc,
                                 slot_mapping, k_out, v_out, cos_sin,
                                 num_q, num_kv, head_dim)
        return (q_out.view(T, num_q, head_dim),
                k_out.view(T, num_kv, head_dim) if want_kv else None,
                v_out.view(T, num_kv, head_dim) if want_kv else None)
    # CPU reference composition (same semantics)
    q_sz, kv_sz = num_q * head_dim, num_kv * head_dim
    q, k, v = qkv.split([q_sz, kv_sz, kv_sz], dim=-1)
    q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
    q, k = rotary_embedding(positions, q, k, cos_sin, head_dim)
    q = q.view(T, num_q, head_dim)
    k = k.view(T, num_kv, head_dim)
    v = v.view(T, num_kv, head_dim)
    if kv_cache is not None:
        if kv_cache[0].dtype == torch.uint8:    # fp8 KV cache
            reshape_and_cache(kv_fp8_quant(k), kv_fp8_quant(v),
                              kv_cache[0], kv_cache[1], slot_mapping)
        else:
            reshape_and_cache(k, v, kv_cache[0], kv_cache[1], slot_mapping)
    return q, (k if want_kv else None), (v if want_kv else None)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        i = x.shape[-1] // 2
        out = torch.empty(*x.shape[:-1], i, dtype=x.dtype, device=x.device)
        _native().silu_and_mul(out, x)
        return out
    return ref.silu_and_mul(x)


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().gelu_tanh(out, x)
        return out
    return ref.gelu_tanh(x)


def reshape_and_cache(k: torch.Tensor, v: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      slot_mapping: torch.Tensor):
    if k.is_cuda:
        _native().reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 cu_seqlens: torch.Tensor, max_seqlen: int,
                 scale: float, causal: bool = True,
                 window: int = 0,
                 cu_seqlens_k: torch.Tensor | None = None) -> torch.Tensor:
    """Varlen attention. When cu_seqlens_k is given, q rows are the
    SUFFIX of each sequence (cached-prefix prefill): kv rows cover the
    full context and causal positions are offset by klen - qlen."""
    if cu_seqlens_k is None:
        cu_seqlens_k = cu_seqlens
    if q.is_cuda:
        out = torch.empty_like(q)
        _native().attn_prefill(out, q, k, v, cu_seqlens, cu_seqlens_k,
                               max_seqlen, scale, causal, window)
        return out
    return ref.attn_prefill(q, k, v, cu_seqlens, max_seqlen, scale, causal,
                            window, cu_seqlens_k)


# Flash-decoding split-K quantum (must match PART_QUANT in
# paged_attn_decode.hip).
DECODE_PARTITION = 128


def decode_workspace(max_batch: int, num_q_heads: int, head_dim: int,
                     max_seq_len: int, device) -> tuple[torch.Tensor, torch.Tensor]:
    """Persistent fp32 scratch for partitioned decode attention."""
    max_parts = max(1, (max_seq_len + DECODE_PARTITION - 1) // DECODE_PARTITION)
    tmp_out = torch.empty(max_batch, num_q_heads, max_parts, head_dim,
                          dtype=torch.float32, device=device)
    tmp_ml = torch.empty(max_batch, num_q_heads, max_parts, 2,
                         dtype=torch.float32, device=device)
    return tmp_out, tmp_ml


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float,
                      workspace=None, max_len: int | None = None,
                      window: int = 0) -> torch.Tensor:
    if q.is_cuda:
        out = torch.empty_like(q)
        if max_len is None:
            max_len = int(seq_lens.max())   # host sync — pass max_len to avoid
        if workspace is None:
            workspace = decode_workspace(q.shape[0], q.shape[1], q.shape[2],
                                         max_len, q.device)
        tmp_out, tmp_ml = workspace
        _native().paged_attn_decode(out, q, k_cache, v_cache, block_tables,
                                    seq_lens, scale,
                                    tmp_out[:q.shape[0]], tmp_ml[:q.shape[0]],
                                    max_len, window)
        return out
    if k_cache.dtype == torch.uint8:            # fp8 KV cache
        k_cache = kv_fp8_dequant(k_cache)
        v_cache = kv_fp8_dequant(v_cache)
    return ref.paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens,
                                 scale, window)


def sample_tokens(logits: torch.Tensor, temperatures: torch.Tensor,
                  seeds: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.int64,
                          device=logits.device)
        _native().sample_tokens(out, logits, temperatures, seeds)
        return out
    return ref.sample_tokens(logits, temperatures, seeds)


def sample_tokens_ext(logits: torch.Tensor, temperatures: torch.Tensor,
                      seeds: torch.Tensor, top_p: torch.Tensor,
                      top_k: torch.Tensor, rep_pen: torch.Tensor,
                      pres_pen: torch.Tensor, freq_pen: torch.Tensor,
                      counts=None, seen=None, row_map=None) -> torch.Tensor:
    """Fused top-k/top-p/penalty Gumbel sampling (GPU only): one block
    per row, histogram-located threshold, no vocab sort. Penalties read
    per-row token-count tables indexed by row_map (engine row slots)."""
    out = torch.empty(logits.shape[0], dtype=torch.int64,
                      device=logits.device)
    _native().sample_tokens_ext(out, logits, temperatures, seeds, top_p,
                                top_k, rep_pen, pres_pen, freq_pen,
                                counts, seen, row_map)
    return out


def quantize_fp8(t: torch.Tensor, dim: int = -1):
    """Per-row (dim=-1 reduces over the last axis) e4m3 quantization.
    Returns (bytes uint8 view, float32 scales)."""
    amax = t.float().abs().amax(dim=dim, keepdim=True).clamp(min=1e-8)
    scale = amax / 448.0                      # e4m3fn max normal
    q = (t.float() / scale).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale.squeeze(dim).float()


def gemm_fp8(x8: torch.Tensor, w8: torch.Tensor, x_scale: torch.Tensor,
             w_scale: torch.Tensor, bias: torch.Tensor | None = None,
             act: int = 0) -> torch.Tensor:
    """out[M,N] = dequant(x8 @ w8^T): e4m3 inputs as uint8 views with
    per-row activation scales and per-output-channel weight scales
    (epilogue dequant; MFMA runs at the 2x fp8 rate)."""
    M, N = x8.shape[0], w8.shape[0]
    if x8.is_cuda:
        out = torch.empty(M, N, dtype=torch.bfloat16, device=x8.device)
        _native().gemm_fp8(out, x8, w8, x_scale.contiguous(),
                           w_scale.contiguous(), bias, act)
        return out
    xf = x8.view(torch.float8_e4m3fn).float() * x_scale[:, None]
    wf = w8.view(torch.float8_e4m3fn).float() * w_scale[:, None]
    out = xf @ wf.t()
    if bias is not None:
        out = out + bias.float()
    if act == 1:
        out = torch.nn.functional.gelu(out, approximate="tanh")
    return out.to(torch.bfloat16)


def gemm_bf16(x: torch.Tensor, w: torch.Tensor, bias=None,
              act: int = 0) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype,
                          device=x.device)
        _native().gemm_bf16(out, x, w, bias, act)
        return out
    return ref.gemm_bf16(x, w, bias, act)


_SKINNY_MAX_M = 1024
_SKINNY_SCRATCH: dict = {}        # device -> fp32 workspace tensor
_SKINNY_SWIZZLE = os.environ.get("HELIX_SKINNY_SWIZZLE", "1") == "1"
_SKINNY_ENABLED = os.environ.get("HELIX_SKINNY_GEMM", "0") == "1"


def _skinny_split(M: int, N: int, K: int) -> int:
    """Split-K factor: enough workgroups to put ~3 on each of the 256
    CUs (m97-structure occupancy), subject to K dividing into 64s."""
    bm = 128 if M > 64 else 64
    tiles = ((M + bm - 1) // bm) * ((N + 127) // 128)
    s = 1
    while s < 8 and tiles * s < 768 and K % (s * 2 * 64) == 0:
        s *= 2
    return s


def gemm_skinny_bf16(x: torch.Tensor, w: torch.Tensor, bias=None,
                     act: int = 0) -> torch.Tensor:
    """Decode-shape GEMM (M <= 1024): split-K skinny tiles, fused
    bias/act epilogue. Caller guarantees bf16 CUDA contiguous inputs."""
    M, K = x.shape
    N = w.shape[0]
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    s = _skinny_split(M, N, K)
    scratch = None
    if s > 1:
        need = s * M * N
        key = x.device.index
        cur = _SKINNY_SCRATCH.get(key)
        if cur is None or cur.numel() < need:
            cur = torch.empty(need, dtype=torch.float32, device=x.device)
            _SKINNY_SCRATCH[key] = cur
        scratch = cur
    _native().gemm_skinny_bf16(out, x, w, bias, scratch, s, act,
                               1 if _SKINNY_SWIZZLE else 0)
    return out


def linear(x: torch.Tensor, w: torch.Tensor, bias=None) -> torch.Tensor:
    """Projection dispatch: skinny-M decode GEMMs on the owned kernel,
    everything else (prefill-sized M, CPU, odd dims) via torch/hipBLASLt."""
    if (_SKINNY_ENABLED and x.is_cuda and x.dtype == torch.bfloat16
            and x.dim() == 2
            and w.dtype == torch.bfloat16 and 0 < x.shape[0] <= _SKINNY_MAX_M
            and x.shape[1] % 64 == 0 and w.shape[0] % 16 == 0
            and x.is_contiguous() and w.is_contiguous()):
        return gemm_skinny_bf16(x, w, bias)
    return torch.nn.functional.linear(x, w, bias)


def mfma_probe(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    d = torch.empty(16, 16, dtype=torch.float32, device=a.device)
    _native().mfma_probe(d, a, b)
    return d
