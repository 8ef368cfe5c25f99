"""Extended randomized soak of the invariant-heavy utilities (beyond the
CI hypothesis budgets): ragged-box tiling and chunked-ring all-reduce.
Run ad hoc: python tools/soak_properties.py [iters]"""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from vescale_amd.checkpoint.ragged_boxes import (
    box_flat_start,
    box_numel,
    break_ragged_box,
)
from vescale_amd.emulator import run_ring_all_reduce


def main(iters: int = 3000) -> None:
    rng = random.Random(0)
    for _ in range(iters):
        nd = rng.randint(1, 4)
        shape = [rng.randint(1, 9) for _ in range(nd)]
        n = 1
        for s in shape:
            n *= s
        a = rng.randint(0, n)
        b = rng.randint(a, n)
        pos = a
        for box in break_ragged_box(shape, a, b):
            assert box_flat_start(shape, box) == pos
            pos += box_numel(box)
        assert pos == b
    for _ in range(max(50, iters // 15)):
        n = rng.randint(1, 5000)
        w = rng.randint(1, 8)
        c = rng.randint(4, 600)
        bufs = [torch.randn(n) for _ in range(w)]
        out = run_ring_all_reduce([x.clone() for x in bufs], chunk_bytes=c * 4)
        ref = sum(x.double() for x in bufs)
        assert torch.allclose(out[0].double(), ref, atol=2e-3)
    print("soak OK")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 3000)
