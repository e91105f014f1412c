"""fp8 (OCP e4m3) KV cache: CPU semantics tests. GPU numerics live in
test_ops_gpu.py / test_engine_gpu.py."""
import pytest
import torch

import helix_amd.ops as ops
from helix_amd.engine.engine import EngineConfig, LLMEngine
from helix_amd.engine.sampling_params import SamplingParams


def test_kv_fp8_roundtrip():
    t = torch.randn(4, 8, dtype=torch.bfloat16)
    q = ops.kv_fp8_quant(t)
    assert q.dtype == torch.uint8
    back = ops.kv_fp8_dequant(q)
    rel = (back.float() - t.float()).norm() / t.float().norm()
    assert rel < 0.06          # e4m3: 3 mantissa bits


def test_decode_with_fp8_cache_matches_dequant_reference():
    torch.manual_seed(0)
    L, hq, hkv, d, bs = 37, 8, 4, 64, 16
    k = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    v = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    q = torch.randn(1, hq, d, dtype=torch.bfloat16)
    nb = (L + bs - 1) // bs
    kc = torch.zeros(nb + 1, hkv, bs, d, dtype=torch.uint8)
    vc = torch.zeros_like(kc)
    slots = torch.arange(bs, bs + L, dtype=torch.int64)
    ops.reshape_and_cache(ops.kv_fp8_quant(k), ops.kv_fp8_quant(v),
                          kc, vc, slots)
    bt = torch.arange(1, nb + 2, dtype=torch.int32).unsqueeze(0)
    lens = torch.tensor([L], dtype=torch.int32)
    scale = d ** -0.5
    out8 = ops.paged_attn_decode(q, kc, vc, bt, lens, scale)
    # identical computation on the dequantized bf16 cache
    out_ref = ops.paged_attn_decode(q, ops.kv_fp8_dequant(kc),
                                    ops.kv_fp8_dequant(vc), bt, lens, scale)
    torch.testing.assert_close(out8, out_ref, atol=1e-3, rtol=1e-3)
    # and close to the full-precision result (quantization error bounded)
    kcf = torch.zeros(nb + 1, hkv, bs, d, dtype=torch.bfloat16)
    vcf = torch.zeros_like(kcf)
    ops.reshape_and_cache(k, v, kcf, vcf, slots)
    out_full = ops.paged_attn_decode(q, kcf, vcf, bt, lens, scale)
    rel = (out8.float() - out_full.float()).norm() / \
        out_full.float().norm()
    assert rel < 0.05, rel


def test_engine_generates_with_fp8_kv():
    cfg = dict(model="tiny-gqa", max_model_len=256, max_num_seqs=4,
               kv_cache_blocks=128, eos_token_id=-1, seed=5)
    sp = SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True)
    prompts = [[1, 2, 3, 4, 5]]
    want = LLMEngine(EngineConfig(**cfg), device="cpu").generate(
        prompts, sp)[0]
    eng = LLMEngine(EngineConfig(**cfg, kv_cache_dtype="fp8"),
                    device="cpu")
    assert eng.kv.caches[0][0].dtype == torch.uint8
    got = eng.generate(prompts, sp)[0]
    assert len(got) == 12
    agree = sum(a == b for a, b in zip(got, want))
    assert agree >= 6, f"fp8-KV diverged early: {got} vs {want}"


def test_fp8_kv_memory_halves():
    from helix_amd.engine.kv_cache import KVCache
    bf16 = KVCache.block_bytes(2, 2, 64, 16, dtype=torch.bfloat16)
    fp8 = KVCache.block_bytes(2, 2, 64, 16, dtype=torch.uint8)
    assert fp8 * 2 == bf16
