"""fa_bwd2 refcheck vs autograd + timing vs library/old bwd (GPU)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, time, torch
import torch.nn.functional as F
import vescale_amd.ops as ops
C = ops.require_ext()

def check(B, Hq, Hkv, S, tag):
    torch.manual_seed(3)
    q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
    sc = 1.0 / math.sqrt(128)
    o, lse = C.fa_fwd(q.detach(), k.detach(), v.detach(), sc)
    fused = os.environ.get("VESCALE_FA_FUSED_DVDK", "1") == "1"
    dq, dk, dv = C.fa_bwd2(q.detach(), k.detach(), v.detach(), o, dy, lse, sc, fused)
    ref = F.scaled_dot_product_attention(q.float(), k.float().repeat_interleave(Hq//Hkv,1),
                                         v.float().repeat_interleave(Hq//Hkv,1), is_causal=True)
    ref.backward(dy.float())
    ok = True
    for name, a, b in (("dq", dq, q.grad), ("dk", dk, k.grad), ("dv", dv, v.grad)):
        e = (a.float() - b.float()).abs().max().item()
        m = b.float().abs().max().item()
        print(f"{tag} {name}: err {e:.4f} refmax {m:.2f} rel {e/m:.4f}")
        ok &= e / m < 0.03
    return ok

ok = check(1, 2, 1, 256, "small")
ok &= check(2, 4, 2, 1024, "mid")
print("REFCHECK", "OK" if ok else "FAILED")
if not ok: raise SystemExit(1)

B, Hq, Hkv, S = 4, 32, 8, 8192
torch.manual_seed(0)
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
dy = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
sc = 1.0 / math.sqrt(128)
o, lse = C.fa_fwd(q, k, v, sc)
def bench(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t)/n
t2 = bench(lambda: C.fa_bwd2(q, k, v, o, dy, lse, sc, True))
print(f"fa_bwd2 fused dvdk: {t2*1e3:.2f} ms")
t2u = bench(lambda: C.fa_bwd2(q, k, v, o, dy, lse, sc, False))
print(f"fa_bwd2 split dvdk: {t2u*1e3:.2f} ms")
t1 = bench(lambda: C.fa_bwd(q, k, v, o, dy, lse, sc))
print(f"fa_bwd (old): {t1*1e3:.2f} ms")
# library backward via autograd on aten path
philox = torch.zeros((), device="cuda", dtype=torch.int64)
def lib():
    torch.ops.aten._scaled_dot_product_flash_attention_backward(
        dy, q, k, v, o, lse, None, None, S, S, 0.0, True, philox, philox, scale=sc)
t3 = bench(lib)
print(f"aotriton bwd: {t3*1e3:.2f} ms")
