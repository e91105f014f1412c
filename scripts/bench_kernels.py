#!/usr/bin/env python3
"""Microbenchmarks for the helix_amd CDNA4 kernels (run on an MI355X).

Times each hot kernel at serving shapes with hip events and prints
bandwidth/TFLOPS vs the roofline. Usage: python scripts/bench_kernels.py
[decode|prefill|gemm|norm|sample|all]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import helix_amd.ops as ops

DEV = "cuda"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def bench_decode():
    print("== paged_attn_decode ==")
    for (B, hq, hkv, d, L) in [(64, 32, 8, 128, 560), (64, 32, 8, 128, 2048),
                               (1, 32, 8, 128, 2048), (8, 32, 8, 128, 8192),
                               (256, 32, 8, 128, 560)]:
        bs = 16
        nb = (L + bs - 1) // bs
        total_blocks = B * nb + 1
        q = torch.randn(B, hq, d, dtype=torch.bfloat16, device=DEV)
        kc = torch.randn(total_blocks, hkv, bs, d, dtype=torch.bfloat16,
                         device=DEV)
        vc = torch.randn_like(kc)
        bt = torch.arange(1, B * nb + 1, dtype=torch.int32,
                          device=DEV).view(B, nb)
        lens = torch.full((B,), L, dtype=torch.int32, device=DEV)
        ws = ops.decode_workspace(B, hq, d, L, DEV)
        scale = d ** -0.5
        t = timeit(lambda: ops.paged_attn_decode(q, kc, vc, bt, lens, scale,
                                                 ws, L))
        bytes_moved = B * hkv * L * d * 2 * 2
        print(f"B={B:4d} L={L:5d}: {t*1e6:8.1f} us  "
              f"{bytes_moved/t/1e12:6.2f} TB/s (peak 6.3)")


def bench_prefill():
    print("== attn_prefill ==")
    for (B, S, hq, hkv, d) in [(64, 512, 32, 8, 128), (8, 2048, 32, 8, 128),
                               (1, 8192, 32, 8, 128)]:
        T = B * S
        q = torch.randn(T, hq, d, dtype=torch.bfloat16, device=DEV)
        k = torch.randn(T, hkv, d, dtype=torch.bfloat16, device=DEV)
        v = torch.randn_like(k)
        cu = torch.arange(0, T + 1, S, dtype=torch.int32, device=DEV)
        scale = d ** -0.5
        t = timeit(lambda: ops.attn_prefill(q, k, v, cu, S, scale), iters=20)
        flops = B * hq * 2 * 2 * d * S * S / 2  # causal half
        print(f"B={B:3d} S={S:5d}: {t*1e3:8.2f} ms  "
              f"{flops/t/1e12:7.1f} TF (dense peak 2500)")



def bench_skinny():
    print("== gemm_skinny_bf16 (split-K, XCD swizzle) vs hipBLASLt ==")
    import helix_amd.ops as _o
    for (M, N, K) in [(512, 6144, 4096), (512, 4096, 4096),
                      (512, 28672, 4096), (512, 4096, 14336),
                      (512, 128256, 4096),
                      (256, 4096, 4096), (64, 6144, 4096),
                      (128, 28672, 4096)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) / 8
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) / 8
        s_k = _o._skinny_split(M, N, K)
        t1 = timeit(lambda: ops.gemm_skinny_bf16(x, w), iters=40)
        t2 = timeit(lambda: torch.nn.functional.linear(x, w), iters=40)
        fl = 2 * M * N * K
        print(f"M{M:5d} N{N:6d} K{K:5d} splitK={s_k}: "
              f"ours {t1*1e6:8.1f} us ({fl/t1/1e12:7.1f} TF) | "
              f"hipblaslt {t2*1e6:8.1f} us ({fl/t2/1e12:7.1f} TF)")


def bench_gemm():
    print("== gemm_bf16 (ours) vs torch/hipBLASLt ==")
    for (M, N, K) in [(512, 3072, 768), (8192, 768, 768), (4096, 4096, 4096),
                      (8192, 8192, 8192), (64, 6144, 4096), (64, 28672, 4096)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV)
        t1 = timeit(lambda: ops.gemm_bf16(x, w), iters=30)
        t2 = timeit(lambda: torch.nn.functional.linear(x, w), iters=30)
        fl = 2 * M * N * K
        print(f"M{M:5d} N{N:5d} K{K:5d}: ours {t1*1e6:9.1f} us "
              f"({fl/t1/1e12:7.1f} TF) | hipblaslt {t2*1e6:9.1f} us "
              f"({fl/t2/1e12:7.1f} TF)")


def bench_fp8():
    print("== gemm_fp8 (MX e4m3, epilogue dequant) vs bf16 paths ==")
    for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (512, 28672, 4096), (512, 6144, 4096),
                      (16384, 4096, 4096)]:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05
        x8, xs = ops.quantize_fp8(x)
        w8, ws = ops.quantize_fp8(w)
        t1 = timeit(lambda: ops.gemm_fp8(x8, w8, xs, ws), iters=30)
        t2 = timeit(lambda: torch.nn.functional.linear(x, w), iters=30)
        fl = 2 * M * N * K
        print(f"M{M:5d} N{N:5d} K{K:5d}: fp8 {t1*1e6:9.1f} us "
              f"({fl/t1/1e12:7.1f} TF) | hipblaslt-bf16 {t2*1e6:9.1f} us "
              f"({fl/t2/1e12:7.1f} TF)")


def bench_norm():
    print("== rms_norm / fused_add_rms_norm / silu / rope ==")
    for T in [64, 8192]:
        H = 4096
        x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
        r = torch.randn_like(x)
        w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: ops.rms_norm(x, w, 1e-5))
        bw = T * H * 2 * 2
        print(f"rms_norm T={T:5d}: {t*1e6:7.1f} us {bw/t/1e12:5.2f} TB/s")
        t = timeit(lambda: ops.fused_add_rms_norm(x, r, w, 1e-5))
        print(f"fused_add T={T:5d}: {t*1e6:7.1f} us {T*H*2*4/t/1e12:5.2f} TB/s")
        g = torch.randn(T, 2 * 14336, dtype=torch.bfloat16, device=DEV)
        t = timeit(lambda: ops.silu_and_mul(g))
        print(f"silu_mul  T={T:5d}: {t*1e6:7.1f} us "
              f"{T*14336*3*2/t/1e12:5.2f} TB/s")


def bench_sample():
    print("== sample_tokens ==")
    for B in [1, 64, 256]:
        V = 128256
        logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV)
        temps = torch.zeros(B, dtype=torch.float32, device=DEV)
        seeds = torch.arange(B, dtype=torch.int64, device=DEV)
        t = timeit(lambda: ops.sample_tokens(logits, temps, seeds))
        print(f"B={B:4d}: {t*1e6:7.1f} us {B*V*2/t/1e12:5.2f} TB/s")
        temps2 = torch.ones(B, dtype=torch.float32, device=DEV)
        t = timeit(lambda: ops.sample_tokens(logits, temps2, seeds))
        print(f"B={B:4d} (gumbel): {t*1e6:7.1f} us")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    torch.manual_seed(0)
    fns = {"decode": bench_decode, "prefill": bench_prefill,
           "gemm": bench_gemm, "skinny": bench_skinny, "fp8": bench_fp8, "norm": bench_norm,
           "sample": bench_sample}
    if which == "all":
        for f in fns.values():
            f()
    else:
        fns[which]()
