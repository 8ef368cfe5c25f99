import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
import vescale_amd.ops as ops
C = ops.require_ext()

def check(M, N, K):
    torch.manual_seed(1)
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) / 8
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) / 8
    ref = a.float() @ b.float().t()
    ok = True
    for mode, name in ((4, "rot3np"), (7, "rot8"), (8, "rot9")):
        # race screen: the rot schedule must be multi-run stable (m152)
        for it in range(3):
            c = C.gemm_tn8(a, b, mode)
            err = (c.float() - ref).abs()
            rel = (err / ref.abs().clamp_min(1e-2)).max().item()
            if rel >= 0.05 or not torch.isfinite(c.float()).all():
                ok = False
            if it == 0:
                c0 = c
            elif not torch.equal(c, c0):
                print(f"  {name} NONDETERMINISTIC at {M}x{N}x{K} run {it}")
                ok = False
        print(f"{M}x{N}x{K} {name}: max_abs={err.max().item():.4f} rel={rel:.4f}")
    return ok

ok = check(512, 512, 256) and check(512, 512, 512) and check(1024, 512, 4096)
print("REFCHECK", "OK" if ok else "FAILED")
if not ok: raise SystemExit(1)

def bench(M, N, K, n=20):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bt = b.t()
    def rot3np(): C.gemm_tn8(a, b, 4)
    def rot9(): C.gemm_tn8(a, b, 8)
    def lib(): torch.matmul(a, bt)
    for fn, name in ((rot3np, "rot3np"), (rot9, "rot9"), (lib, "hipblaslt")):
        for _ in range(3): fn()
        torch.cuda.synchronize(); t = time.perf_counter()
        for _ in range(n): fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t) / n
        print(f"{M}x{N}x{K} {name}: {dt*1e3:7.3f} ms {2*M*N*K/dt/1e12:6.0f} TF")

bench(4096, 4096, 4096)
bench(8192, 8192, 8192)
bench(32768, 6144, 4096)   # llama wqkv
bench(32768, 4096, 14336)  # llama w2 fwd
