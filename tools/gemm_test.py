import torch, time, sys
import os
VARIANT = int(os.environ.get("V", "0"))
import vescale_amd.ops as ops
C = ops.require_ext()

def refcheck(M, N, K, tag=""):
    torch.manual_seed(0)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    got = C.gemm_tn(a, b, VARIANT)
    ref = (a.float() @ b.float().t())
    gf = got.float()
    ok = torch.allclose(gf, ref, atol=2.0, rtol=2e-2)
    rel = ((gf - ref).abs() / (ref.abs() + 1)).max().item()
    print(f"refcheck {tag} M{M} N{N} K{K}: ok={ok} maxrel={rel:.4f}", flush=True)
    if not ok:
        bad = ((gf - ref).abs() / (ref.abs()+1)) > 0.05
        print("  bad frac:", bad.float().mean().item())
        idx = bad.nonzero()[:5]
        for i in idx:
            r, c = int(i[0]), int(i[1])
            print(f"  C[{r},{c}] got {gf[r,c]:.3f} ref {ref[r,c]:.3f}")
    return ok

# identity check with asymmetric B (guide G9: transpose-detecting)
def idcheck():
    M, N, K = 512, 256, 512
    a = torch.zeros(M, K, device="cuda", dtype=torch.bfloat16)
    for i in range(min(M, K)):
        a[i, i] = 1.0
    b = (torch.arange(N, device="cuda").view(N,1) * 1000 + torch.arange(K, device="cuda").view(1,K)).bfloat16() * 0.001
    got = C.gemm_tn(a, b, VARIANT).float()
    ref = (a.float() @ b.float().t())
    ok = torch.allclose(got, ref, atol=0.5)
    print("identity-asym check:", ok, flush=True)
    if not ok:
        print(got[:3,:5]); print(ref[:3,:5])
    return ok

def bench(M, N, K, tag):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(3): C.gemm_tn(a, b, VARIANT)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): C.gemm_tn(a, b, VARIANT)
    torch.cuda.synchronize(); t1 = time.perf_counter()
    tf = 2*M*N*K*10/(t1-t0)/1e12
    # library comparison
    bt = b.t().contiguous().t()
    for _ in range(3): a @ b.t()
    torch.cuda.synchronize(); t2 = time.perf_counter()
    for _ in range(10): a @ b.t()
    torch.cuda.synchronize(); t3 = time.perf_counter()
    tf_lib = 2*M*N*K*10/(t3-t2)/1e12
    print(f"bench {tag}: ours {tf:.0f} TF vs lib {tf_lib:.0f} TF", flush=True)

ok = idcheck()
pass_tiny = True
ok &= refcheck(512, 256, 4096, "1tile")
ok &= refcheck(1024, 512, 4096)
ok &= refcheck(4096, 4096, 4096)
if ok:
    bench(4096, 4096, 4096, "4k3")
    bench(16384, 28672, 4096, "w13 fwd")
    bench(16384, 4096, 14336, "w2 fwd")
    bench(16384, 128256, 4096, "lmhead")
else:
    sys.exit(1)
