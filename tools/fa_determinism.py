import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
import vescale_amd.ops as ops
C = ops.require_ext()
B, Hq, Hkv, S = 4, 32, 8, 8192
torch.manual_seed(0)
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
dy = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
sc = 1.0 / math.sqrt(128)
o0, l0 = C.fa_fwd(q, k, v, sc)
for i in range(5):
    o1, l1 = C.fa_fwd(q, k, v, sc)
    neq = (o1 != o0).sum().item(); lneq = (l1 != l0).sum().item()
    print(f"fwd run {i}: o diff elems {neq}, lse diff {lneq}")
g0 = C.fa_bwd2(q, k, v, o0, dy, l0, sc, True)
for i in range(3):
    g1 = C.fa_bwd2(q, k, v, o0, dy, l0, sc, True)
    print(f"bwd run {i}: diffs", [ (a != b).sum().item() for a, b in zip(g0, g1) ])
