import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
import torch.nn.functional as F
import vescale_amd.ops as ops
C = ops.require_ext()

B, Hq, Hkv, S = 1, 2, 1, 256
torch.manual_seed(17)
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
sc = 1.0 / math.sqrt(128)
r = F.scaled_dot_product_attention(
    q.float().repeat_interleave(1, 1), k.float().repeat_interleave(Hq // Hkv, 1),
    v.float().repeat_interleave(Hq // Hkv, 1), is_causal=True)
for mode in (0, 4, 5):
    o, lse = C.fa_fwd_ablate(q, k, v, sc, mode)
    err = (o.float() - r).abs()
    print(f"mode {mode}: max_abs={err.max().item():.4f}")
    if err.max().item() > 0.05:
        # localize: per 32-q-row block and per 32-d block
        eq = err.amax(dim=(0, 1, 3)).reshape(-1, 32).amax(1)
        ed = err.amax(dim=(0, 1, 2)).reshape(-1, 32).amax(1)
        print("  err by q-block32:", [f"{x:.2f}" for x in eq.tolist()])
        print("  err by d-block32:", [f"{x:.2f}" for x in ed.tolist()])
