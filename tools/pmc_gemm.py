import os, torch
import vescale_amd.ops as ops
C = ops.require_ext()
a = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
b = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
V = int(os.environ.get("V", "0"))
for _ in range(5):
    C.gemm_tn(a, b, V)
torch.cuda.synchronize()
