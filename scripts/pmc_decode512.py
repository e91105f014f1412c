import sys, torch
sys.path.insert(0, "/root/repo")
import helix_amd.ops as ops
# decode-attention shape at the headline config: B=512, ctx ~528
B, hq, hkv, d, bs, L = 512, 32, 8, 128, 16, 528
nb = (L + bs - 1) // bs
q = torch.randn(B, hq, d, dtype=torch.bfloat16, device="cuda")
kc = torch.randn(B * nb + 1, hkv, bs, d, dtype=torch.bfloat16, device="cuda")
vc = torch.randn_like(kc)
bt = (torch.arange(B * nb, dtype=torch.int32, device="cuda")
      .reshape(B, nb) + 1)
lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
ws = ops.decode_workspace(B, hq, d, 8192, torch.device("cuda"))
for _ in range(10):
    ops.paged_attn_decode(q, kc, vc, bt, lens, d ** -0.5, ws, L)
torch.cuda.synchronize()
print("done")
