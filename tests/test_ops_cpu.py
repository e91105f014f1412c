"""CPU sanity tests for the reference op implementations (numerics contract).

These pin the semantics the HIP kernels are tested against.
"""
import math

import torch

import helix_amd.ops as ops
import helix_amd.ops.reference as ref


def test_rms_norm_known_value():
    x = torch.full((1, 8), 2.0, dtype=torch.bfloat16)
    w = torch.ones(8, dtype=torch.bfloat16)
    out = ops.rms_norm(x, w, 0.0)
    assert torch.allclose(out.float(), torch.ones(1, 8), atol=1e-2)


def test_fused_add_rms_norm_semantics():
    x = torch.randn(4, 16, dtype=torch.bfloat16)
    r = torch.randn(4, 16, dtype=torch.bfloat16)
    w = torch.ones(16, dtype=torch.bfloat16)
    x2, r2 = x.clone(), r.clone()
    out, new_res = ops.fused_add_rms_norm(x2, r2, w, 1e-5)
    assert torch.allclose(new_res.float(), (x.float() + r.float()), atol=2e-2)
    assert torch.allclose(out.float(), ref.rms_norm(new_res, w, 1e-5).float(),
                          atol=1e-2)


def test_rope_preserves_norm():
    d = 64
    q = torch.randn(5, 4 * d, dtype=torch.bfloat16)
    k = torch.randn(5, 2 * d, dtype=torch.bfloat16)
    pos = torch.arange(5, dtype=torch.int64)
    cs = ref.make_cos_sin_cache(d, 16)
    q2, k2 = ops.rotary_embedding(pos, q.clone(), k.clone(), cs, d)
    # rotation preserves the per-head L2 norm
    for h in range(4):
        a = q.float().view(5, 4, d)[:, h].norm(dim=-1)
        b = q2.float().view(5, 4, d)[:, h].norm(dim=-1)
        assert torch.allclose(a, b, rtol=2e-2)
    # position 0 is identity
    assert torch.allclose(q2[0].float(), q[0].float(), atol=1e-2)


def test_silu_and_mul():
    x = torch.cat([torch.zeros(3, 8), torch.randn(3, 8)], dim=-1).bfloat16()
    out = ops.silu_and_mul(x)
    assert torch.allclose(out.float(), torch.zeros(3, 8))  # silu(0)=0


def test_attn_prefill_causal_first_token():
    """First token attends only to itself => out == v[0]."""
    q = torch.randn(7, 4, 64, dtype=torch.bfloat16)
    k = torch.randn(7, 2, 64, dtype=torch.bfloat16)
    v = torch.randn(7, 2, 64, dtype=torch.bfloat16)
    cu = torch.tensor([0, 7], dtype=torch.int32)
    out = ops.attn_prefill(q, k, v, cu, 7, 0.125)
    want0 = v[0].repeat_interleave(2, dim=0)
    assert torch.allclose(out[0].float(), want0.float(), atol=2e-2)


def test_paged_decode_matches_prefill_last_row():
    """Decode of the last position must equal prefill's last-row output."""
    torch.manual_seed(0)
    L, hq, hkv, d, bs = 37, 8, 4, 64, 16
    q = torch.randn(L, hq, d, dtype=torch.bfloat16)
    k = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    v = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    cu = torch.tensor([0, L], dtype=torch.int32)
    scale = d ** -0.5
    pre = ops.attn_prefill(q, k, v, cu, L, scale)

    nb = (L + bs - 1) // bs
    kc = torch.zeros(nb + 1, hkv, bs, d, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    slots = torch.arange(bs, bs + L, dtype=torch.int64)  # blocks 1..
    ops.reshape_and_cache(k, v, kc, vc, slots)
    bt = torch.arange(1, nb + 2, dtype=torch.int32).unsqueeze(0)
    out = ops.paged_attn_decode(q[-1:].reshape(1, hq, d), kc, vc, bt,
                                torch.tensor([L], dtype=torch.int32), scale)
    assert torch.allclose(out[0].float(), pre[-1].float(), atol=3e-2,
                          rtol=3e-2)


def test_sample_greedy_cpu():
    logits = torch.randn(4, 100, dtype=torch.bfloat16)
    temps = torch.zeros(4)
    seeds = torch.zeros(4, dtype=torch.int64)
    out = ops.sample_tokens(logits, temps, seeds)
    assert torch.equal(out, logits.float().argmax(-1))


def test_gemm_reference():
    x = torch.randn(5, 64, dtype=torch.bfloat16)
    w = torch.randn(7, 64, dtype=torch.bfloat16)
    out = ops.gemm_bf16(x, w)
    assert torch.allclose(out.float(), (x.float() @ w.float().t()), atol=0.5)


def test_cos_sin_cache():
    cs = ref.make_cos_sin_cache(8, 4, base=10000.0)
    assert cs.shape == (4, 8)
    assert torch.allclose(cs[0, :4], torch.ones(4))  # cos(0)
    assert torch.allclose(cs[0, 4:], torch.zeros(4))  # sin(0)
    assert math.isclose(float(cs[1, 0]), math.cos(1.0), rel_tol=1e-5)


def test_sliding_window_reference_consistency():
    """Windowed decode of the last position == windowed prefill last row."""
    torch.manual_seed(5)
    L, hq, hkv, d, bs, W = 50, 8, 4, 64, 16, 16
    q = torch.randn(L, hq, d, dtype=torch.bfloat16)
    k = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    v = torch.randn(L, hkv, d, dtype=torch.bfloat16)
    cu = torch.tensor([0, L], dtype=torch.int32)
    scale = d ** -0.5
    pre = ops.attn_prefill(q, k, v, cu, L, scale, window=W)
    nb = (L + bs - 1) // bs
    kc = torch.zeros(nb + 1, hkv, bs, d, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    ops.reshape_and_cache(k, v, kc, vc,
                          torch.arange(bs, bs + L, dtype=torch.int64))
    bt = torch.arange(1, nb + 2, dtype=torch.int32).unsqueeze(0)
    out = ops.paged_attn_decode(q[-1:].reshape(1, hq, d), kc, vc, bt,
                                torch.tensor([L], dtype=torch.int32), scale,
                                window=W)
    assert torch.allclose(out[0].float(), pre[-1].float(), atol=3e-2,
                          rtol=3e-2)
    # and differs from full attention (the window actually applies)
    full = ops.attn_prefill(q, k, v, cu, L, scale)
    assert not torch.allclose(full[-1].float(), pre[-1].float(), atol=1e-3)
