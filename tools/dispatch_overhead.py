"""Measure DTensor dispatch overhead vs plain torch (the
reference treats per-op python dispatch as the eager hot loop's CPU
budget, ~µs per op).  Single rank, world_size 1, CPU: isolates OUR
dispatcher cost (rule lookup + spec cache + wrap) from kernel time.

Usage: python tools/dispatch_overhead.py [iters]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.distributed as dist


def bench(fn, iters):
    for _ in range(100):
        fn()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters * 1e6  # µs/op


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 3000
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29719")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group("gloo", rank=0, world_size=1)
    from vescale_amd.dtensor import Replicate, Shard, distribute_tensor, init_device_mesh

    mesh = init_device_mesh("cpu", (1,))
    t = torch.randn(64, 64)
    d = distribute_tensor(t.clone(), mesh, [Shard(0)])
    w = torch.randn(64, 64)
    dw = distribute_tensor(w.clone(), mesh, [Replicate()])

    cases = {
        "add_":    (lambda: t.add(1.0),          lambda: d.add(1.0)),
        "mul":     (lambda: t * t,               lambda: d * d),
        "matmul":  (lambda: t @ w,               lambda: d @ dw),
        "sum":     (lambda: t.sum(),             lambda: d.sum()),
        "softmax": (lambda: t.softmax(-1),       lambda: d.softmax(-1)),
        "view":    (lambda: t.reshape(-1, 128),  lambda: d.reshape(-1, 128)),
    }
    print(f"{'op':>8} {'torch µs':>9} {'dtensor µs':>11} {'overhead µs':>12}")
    for name, (plain, dt) in cases.items():
        a = bench(plain, iters)
        b = bench(dt, iters)
        print(f"{name:>8} {a:>9.2f} {b:>11.2f} {b - a:>12.2f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
