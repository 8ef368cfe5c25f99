import os, math, time, torch
import torch.nn.functional as F
import vescale_amd.ops as ops
C = ops.require_ext()
from vescale_amd.ops import flash_attention_causal

def refcheck(B, Hq, Hkv, S, tag=""):
    torch.manual_seed(0)
    D = 128
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    out = flash_attention_causal(q, k, v)
    out.backward(dy)
    gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
    q.grad = k.grad = v.grad = None
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=Hq != Hkv)
    ref.backward(dy)
    ok = True
    for name, a, b in [("out", out, ref), ("dq", gq, q.grad), ("dk", gk, k.grad), ("dv", gv, v.grad)]:
        close = torch.allclose(a.float(), b.float(), atol=5e-2, rtol=5e-2)
        err = (a.float() - b.float()).abs().max().item()
        rel = ((a.float()-b.float()).abs()/(b.float().abs()+0.1)).max().item()
        print(f"  {tag} {name}: close={close} maxabs={err:.4f} maxrel={rel:.4f}", flush=True)
        ok &= close
    return ok

def bench(B, Hq, Hkv, S):
    D = 128
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    def run(fn):
        for _ in range(2):
            o = fn(); o.backward(dy); q.grad=k.grad=v.grad=None
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(5):
            o = fn(); o.backward(dy); q.grad=k.grad=v.grad=None
        torch.cuda.synchronize(); return (time.perf_counter()-t0)/5*1e3
    ours = run(lambda: flash_attention_causal(q, k, v))
    ref = run(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=Hq != Hkv))
    print(f"bench B{B} Hq{Hq} S{S}: ours {ours:.2f} ms vs aotriton {ref:.2f} ms (fwd+bwd)", flush=True)

ok = refcheck(1, 1, 1, 128, "tiny")
ok &= refcheck(1, 2, 2, 256, "mha")
ok &= refcheck(2, 8, 2, 512, "gqa")
ok &= refcheck(1, 32, 8, 2048, "llama-shape")
print("ALL OK" if ok else "FAIL", flush=True)
if ok:
    bench(2, 32, 8, 8192)
