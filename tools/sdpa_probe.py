import torch, time
import torch.nn.functional as F

def bench_sdpa(tag):
    B,H,Hk,S,D = 2,32,8,8192,128
    q = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B,Hk,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B,Hk,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    def run():
        o = F.scaled_dot_product_attention(q,k,v,is_causal=True,enable_gqa=True)
        o.backward(torch.ones_like(o))
        q.grad=k.grad=v.grad=None
    for _ in range(2): run()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(5): run()
    torch.cuda.synchronize(); t1=time.perf_counter()
    # fwd only
    with torch.no_grad():
        qq=q.detach()
        torch.cuda.synchronize(); t2=time.perf_counter()
        for _ in range(5):
            F.scaled_dot_product_attention(qq,k.detach(),v.detach(),is_causal=True,enable_gqa=True)
        torch.cuda.synchronize(); t3=time.perf_counter()
    print(f"{tag}: fwd+bwd {(t1-t0)/5*1e3:.1f} ms, fwd {(t3-t2)/5*1e3:.1f} ms")

print("backends: flash", torch.backends.cuda.flash_sdp_enabled(), "mem_eff", torch.backends.cuda.mem_efficient_sdp_enabled(), "math", torch.backends.cuda.math_sdp_enabled())
try:
    print("fa lib:", torch.backends.cuda.preferred_rocm_fa_library())
except Exception as e:
    print("no fa lib api:", e)
bench_sdpa("default")
try:
    torch.backends.cuda.preferred_rocm_fa_library("ck")
    bench_sdpa("ck")
except Exception as e:
    print("ck unavailable:", e)
try:
    from torch.nn.attention import sdpa_kernel, SDPBackend
    with sdpa_kernel(SDPBackend.EFFICIENT_ATTENTION):
        bench_sdpa("mem_efficient")
except Exception as e:
    print("mem_eff fail:", e)
