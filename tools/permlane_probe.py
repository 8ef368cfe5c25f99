"""Verify permlane32_swap semantics: which result element carries the
partner half-wave's value (run on GPU)."""
import torch
import vescale_amd.ops as ops
C = ops.require_ext()
out = C.permlane_probe()
ids = out.cpu().tolist()
# out[l] = (r0, r1) packed; input was lane id
for l in (0, 1, 31, 32, 63):
    r0, r1 = ids[l] >> 16, ids[l] & 0xFFFF
    print(f"lane {l}: r0={r0} r1={r1}")
