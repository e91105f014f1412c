import os, sys, torch
sys.path.insert(0, "/root/repo")
import helix_amd.ops as ops
B, S, hq, hkv, d = 32, 1024, 32, 8, 128
T = B * S
q = torch.randn(T, hq, d, dtype=torch.bfloat16, device="cuda")
k = torch.randn(T, hkv, d, dtype=torch.bfloat16, device="cuda")
v = torch.randn_like(k)
cu = torch.arange(0, T + 1, S, dtype=torch.int32, device="cuda")
for _ in range(10):
    ops.attn_prefill(q, k, v, cu, S, d ** -0.5)
torch.cuda.synchronize()
print("done")
