#!/usr/bin/env python3
"""A/B: decode attention with bf16 vs fp8 (e4m3) KV cache."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

import helix_amd.ops as ops  # noqa: E402

for (B, L) in [(64, 2048), (256, 560), (64, 560), (8, 8192)]:
    hq, hkv, d, bs = 32, 8, 128, 16
    nb = (L + bs - 1) // bs
    q = torch.randn(B, hq, d, dtype=torch.bfloat16, device="cuda")
    bt = (torch.arange(B * nb, dtype=torch.int32, device="cuda")
          .reshape(B, nb) + 1)
    lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
    ws = ops.decode_workspace(B, hq, d, L, torch.device("cuda"))
    res = {}
    for tag in ("bf16", "fp8"):
        kc = (torch.randn(B * nb + 1, hkv, bs, d, device="cuda") * 0.5) \
            .to(torch.bfloat16)
        vc = torch.randn_like(kc)
        if tag == "fp8":
            kc = ops.kv_fp8_quant(kc).contiguous()
            vc = ops.kv_fp8_quant(vc).contiguous()
        for _ in range(5):
            ops.paged_attn_decode(q, kc, vc, bt, lens, d ** -0.5, ws, L)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            ops.paged_attn_decode(q, kc, vc, bt, lens, d ** -0.5, ws, L)
        torch.cuda.synchronize()
        res[tag] = (time.perf_counter() - t0) / 30
    eb = B * hkv * L * d * 2 * 2  # bf16-equivalent bytes
    print(f"B={B:4d} L={L:5d}: bf16 {res['bf16']*1e6:7.1f} us "
          f"({eb/res['bf16']/1e12:4.2f} TB/s-eq) | "
          f"fp8 {res['fp8']*1e6:7.1f} us  "
          f"speedup {res['bf16']/res['fp8']:.2f}x")
