"""fa_fwd ablation timing at llama-8B shape (GPU)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, time, torch
import torch.nn.functional as F
import vescale_amd.ops as ops
C = ops.require_ext()

B, Hq, Hkv, S = 4, 32, 8, 8192
q = torch.randn(B, Hq, S, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, 128, device="cuda", dtype=torch.bfloat16)
sc = 1.0 / math.sqrt(128)
flops = 2 * 2 * B * Hq * (S * S / 2) * 128

def bench(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / n

names = {0: "full", 1: "no-softmax", 2: "no-PV", 3: "no-ST"}
for mode in (0, 1, 2, 3):
    t = bench(lambda m=mode: C.fa_fwd_ablate(q, k, v, sc, m))
    print(f"mode {mode} ({names[mode]:>10}): {t*1e3:7.3f} ms  {flops/t/1e12:5.0f} TF-equiv")
t_lib = bench(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True))
print(f"aotriton fwd        : {t_lib*1e3:7.3f} ms  {flops/t_lib/1e12:5.0f} TF")
