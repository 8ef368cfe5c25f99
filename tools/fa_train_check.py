"""fa_fwd + aten-backward training-path check + timing (GPU)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, time, torch
import torch.nn.functional as F
from vescale_amd.ops.functional import flash_attention_causal

B, Hq, Hkv, S = 2, 4, 2, 512
torch.manual_seed(3)
def mk(*shape):
    return torch.randn(*shape, device="cuda", dtype=torch.bfloat16, requires_grad=True)
q, k, v = mk(B, Hq, S, 128), mk(B, Hkv, S, 128), mk(B, Hkv, S, 128)
o = flash_attention_causal(q, k, v)
g = torch.randn_like(o)
o.backward(g)
gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
q2 = q.detach().float().requires_grad_(); k2 = k.detach().float().requires_grad_()
v2 = v.detach().float().requires_grad_()
r = F.scaled_dot_product_attention(q2, k2, v2, is_causal=True, enable_gqa=True)
r.backward(g.float())
for name, a, b in (("dq", gq, q2.grad), ("dk", gk, k2.grad), ("dv", gv, v2.grad)):
    e = (a.float() - b).abs().max().item()
    m = b.abs().max().item()
    print(f"{name}: max_abs_err {e:.4f} (ref max {m:.2f}) rel {e/m:.4f}")

# timing fwd+bwd at llama shape, ours vs library
B, Hq, Hkv, S = 4, 32, 8, 8192
q = mk(B, Hq, S, 128); k = mk(B, Hkv, S, 128); v = mk(B, Hkv, S, 128)
def step_ours():
    o = flash_attention_causal(q, k, v)
    o.backward(torch.ones_like(o))
    q.grad = k.grad = v.grad = None
def step_lib():
    o = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
    o.backward(torch.ones_like(o))
    q.grad = k.grad = v.grad = None
def bench(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t)/n
print(f"ours fwd+bwd: {bench(step_ours)*1e3:.2f} ms")
print(f"lib  fwd+bwd: {bench(step_lib)*1e3:.2f} ms")
