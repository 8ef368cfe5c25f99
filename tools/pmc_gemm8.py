import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import vescale_amd.ops as ops
C = ops.require_ext()
M, N, K = (int(x) for x in os.environ.get("SHAPE", "8192,8192,8192").split(","))
a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
MODE = int(os.environ.get("MODE", "2"))
for _ in range(8):
    C.gemm_tn8(a, b, MODE)
torch.cuda.synchronize()
