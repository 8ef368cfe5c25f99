import os, sys
sys.path.insert(0, "/root/repo")
import math, time, torch
import torch.nn.functional as F
import vescale_amd.ops as ops
C = ops.require_ext()
B, Hq, Hkv, S = 1, 2, 1, 256
torch.manual_seed(17)
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
sc = 1.0 / math.sqrt(128)
r = F.scaled_dot_product_attention(q.float(), k.float().repeat_interleave(2,1),
                                   v.float().repeat_interleave(2,1), is_causal=True)
o1, _ = C.fa_fwd(q, k, v, sc)
o2, _ = C.fa_fwd_ablate(q, k, v, sc, 0)
print("fa_fwd    err:", (o1.float()-r).abs().max().item())
print("ablate0   err:", (o2.float()-r).abs().max().item())
B, Hq, Hkv, S = 4, 32, 8, 8192
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
def bench(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t)/n
print("fa_fwd  big ms:", bench(lambda: C.fa_fwd(q, k, v, sc))*1e3)
print("ablate0 big ms:", bench(lambda: C.fa_fwd_ablate(q, k, v, sc, 0))*1e3)
o1, _ = C.fa_fwd(q, k, v, sc)
rr = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
print("big rel err:", ((o1.float()-rr.float()).abs()/(rr.float().abs().clamp_min(1))).max().item())

