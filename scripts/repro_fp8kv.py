import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import helix_amd.ops as ops
B, hq, hkv, d, bs, L = 4, 8, 2, 128, 16, 77
nb = (L + bs - 1) // bs
q = torch.randn(B, hq, d, dtype=torch.bfloat16, device="cuda")
kc = torch.zeros(B * nb + 1, hkv, bs, d, dtype=torch.uint8, device="cuda")
vc = torch.zeros_like(kc)
bt = (torch.arange(B * nb, dtype=torch.int32, device="cuda").reshape(B, nb) + 1)
lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
print("A: bf16 decode first")
kc16 = torch.zeros(B * nb + 1, hkv, bs, d, dtype=torch.bfloat16, device="cuda")
vc16 = torch.zeros_like(kc16)
o = ops.paged_attn_decode(q, kc16, vc16, bt, lens, d ** -0.5)
torch.cuda.synchronize(); print("bf16 ok", o.shape)
print("B: fp8 decode")
o = ops.paged_attn_decode(q, kc, vc, bt, lens, d ** -0.5)
torch.cuda.synchronize(); print("fp8 ok", o.shape)
print("C: quantize path")
k = torch.randn(L, hkv, d, dtype=torch.bfloat16, device="cuda")
q8 = ops.kv_fp8_quant(k)
torch.cuda.synchronize(); print("quant ok", q8.dtype, q8.shape)
d16 = ops.kv_fp8_dequant(q8)
torch.cuda.synchronize(); print("dequant ok", d16.dtype)
