import os, sys, torch
sys.path.insert(0, "/root/repo")
import helix_amd.ops as ops
B, hq, hkv, d, L, bs = 256, 32, 8, 128, 560, 16
nb = (L + bs - 1) // bs
q = torch.randn(B, hq, d, dtype=torch.bfloat16, device="cuda")
kc = torch.randn(B*nb+1, hkv, bs, d, dtype=torch.bfloat16, device="cuda")
vc = torch.randn_like(kc)
bt = torch.arange(1, B*nb+1, dtype=torch.int32, device="cuda").view(B, nb)
lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
ws = ops.decode_workspace(B, hq, d, L, "cuda")
for _ in range(20):
    ops.paged_attn_decode(q, kc, vc, bt, lens, d**-0.5, ws, L)
torch.cuda.synchronize()
print("done")
