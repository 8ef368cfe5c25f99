import torch, time
torch.manual_seed(0)
def bench(M, N, K, tag, ta=False, tb=False):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
    if ta: a = a.t().contiguous().t()
    if tb: b = b.t().contiguous().t()
    for _ in range(3): c = a @ b
    torch.cuda.synchronize(); t0 = time.perf_counter()
    n = 10
    for _ in range(n): c = a @ b
    torch.cuda.synchronize(); t1 = time.perf_counter()
    tf = 2*M*N*K*n/(t1-t0)/1e12
    print(f"{tag:28s} M={M:6d} N={N:6d} K={K:6d} ta={int(ta)} tb={int(tb)}: {tf:7.0f} TF  {(t1-t0)/n*1e3:7.2f} ms")

T = 16384  # tokens (bs2 x 8192)
# forward
bench(T, 6144, 4096,  "fwd wqkv")
bench(T, 4096, 4096,  "fwd wo")
bench(T, 28672, 4096, "fwd w13")
bench(T, 4096, 14336, "fwd w2")
bench(T, 128256, 4096,"fwd lmhead")
# dgrad (dy @ W): same shapes transposed
bench(T, 4096, 6144,  "dgrad wqkv")
bench(T, 14336, 4096, "dgrad w2")
bench(T, 4096, 28672, "dgrad w13")
bench(T, 4096, 128256,"dgrad lmhead")
# wgrad (dy^T @ x): M=out_features N=in K=T, A transposed view
def bench_wgrad(OUT, IN, tag):
    dy = torch.randn(T, OUT, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(T, IN, device="cuda", dtype=torch.bfloat16)
    for _ in range(3): g = dy.t() @ x
    torch.cuda.synchronize(); t0 = time.perf_counter()
    n = 10
    for _ in range(n): g = dy.t() @ x
    torch.cuda.synchronize(); t1 = time.perf_counter()
    tf = 2*OUT*IN*T*n/(t1-t0)/1e12
    print(f"{tag:28s} OUT={OUT:6d} IN={IN:6d} K={T}: {tf:7.0f} TF  {(t1-t0)/n*1e3:7.2f} ms")
bench_wgrad(6144, 4096, "wgrad wqkv")
bench_wgrad(4096, 4096, "wgrad wo")
bench_wgrad(28672, 4096,"wgrad w13")
bench_wgrad(4096, 14336,"wgrad w2")
bench_wgrad(128256, 4096,"wgrad lmhead")
