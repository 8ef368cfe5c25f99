"""Offline TunableOp probe: tune hipBLASLt algo selection for the bench's
GEMM shapes in isolation (the in-model tuning run OOMs at 280 GB), then
time tuned vs default.  If it pays, ship the CSV + enable in bench."""
import os, sys, time
import torch

SHAPES = [  # (M, N, K) of y = x[M,K] @ w[N,K]^T per layer GEMM
    (32768, 6144, 4096),    # wqkv
    (32768, 4096, 4096),    # wo
    (32768, 28672, 4096),   # w13
    (32768, 4096, 14336),   # w2
    (32768, 128256, 4096),  # lm head
]

def run_all():
    t = 0.0
    for (M, N, K) in SHAPES:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        for _ in range(3):
            y = x @ w.t(); dx = dy @ w; dw = dy.t() @ x
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            y = x @ w.t(); dx = dy @ w; dw = dy.t() @ x
        torch.cuda.synchronize()
        t += (time.perf_counter() - t0) / 10
        del x, w, dy, y, dx, dw
        torch.cuda.empty_cache()
    return t

mode = sys.argv[1] if len(sys.argv) > 1 else "default"
if mode == "tune":
    # cap per-candidate cost so the 18 ms lm-head GEMMs don't blow the box
    # budget (defaults: 30 ms / 100 iters per candidate)
    torch.cuda.tunable.enable(True)
    torch.cuda.tunable.tuning_enable(True)
    torch.cuda.tunable.set_max_tuning_duration(60)
    torch.cuda.tunable.set_max_tuning_iterations(3)
    torch.cuda.tunable.set_filename("gpurun_out/tunableop_gfx950.csv")
    if os.path.exists("gpurun_out/tunableop_gfx950.csv"):
        torch.cuda.tunable.read_file()
    run_all()
    torch.cuda.tunable.write_file()
    torch.cuda.tunable.tuning_enable(False)
    print("tuned:", run_all() * 1e3, "ms (all shapes fwd+dgrad+wgrad)")
elif mode == "replay":
    torch.cuda.tunable.enable(True)
    torch.cuda.tunable.tuning_enable(False)
    torch.cuda.tunable.set_filename("gpurun_out/tunableop_gfx950.csv")
    torch.cuda.tunable.read_file()
    print("replay:", run_all() * 1e3, "ms (all shapes fwd+dgrad+wgrad)")
else:
    print("default:", run_all() * 1e3, "ms (all shapes fwd+dgrad+wgrad)")
