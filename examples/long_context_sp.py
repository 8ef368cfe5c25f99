"""Long-context sequence parallelism: Ulysses vs ring attention.

A toy causal transformer block runs with its sequence dimension sharded
across ranks; attention switches between the two exact SP algorithms:

  --algo ulysses : Shard(seq)<->Shard(head) all-to-alls, local SDPA
  --algo ring    : KV circulates the ring, log-sum-exp merge, O(S/p) KV

Run (CPU smoke):
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/long_context_sp.py --algo ring --steps 3
On MI355X the same script runs one rank per GPU over RCCL (xGMI:
Ulysses' a2a uses all 7 links; ring's hops use one link each).

Every rank prints the same loss as a single-device run of the same model
(seeded identically) — the SP algorithms are exact, not approximations.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F


class SPBlock(nn.Module):
    """Pre-norm attention+MLP block; attention core is SP-pluggable."""

    def __init__(self, dim, heads, attn):
        super().__init__()
        self.h, self.d = heads, dim // heads
        self.norm1 = nn.LayerNorm(dim)
        self.qkv = nn.Linear(dim, 3 * dim, bias=False)
        self.proj = nn.Linear(dim, dim, bias=False)
        self.norm2 = nn.LayerNorm(dim)
        self.mlp = nn.Sequential(nn.Linear(dim, 4 * dim), nn.GELU(), nn.Linear(4 * dim, dim))
        self.attn = attn

    def forward(self, x):  # x: [B, S_local, dim] (plain local tensor)
        b, s, _ = x.shape
        qkv = self.qkv(self.norm1(x)).reshape(b, s, 3, self.h, self.d)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # [B,H,S,D]
        o = self.attn(q, k, v)  # SP attention over the sharded seq dim
        o = o.transpose(1, 2).reshape(b, s, -1)
        x = x + self.proj(o)
        return x + self.mlp(self.norm2(x))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--algo", default="ring", choices=["ulysses", "ring"])
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--dim", type=int, default=64)
    ap.add_argument("--heads", type=int, default=4)
    ap.add_argument("--seq", type=int, default=64)
    args = ap.parse_args()

    backend = os.environ.get("VESCALE_BACKEND", "nccl" if torch.cuda.is_available() else "gloo")
    dist.init_process_group(backend)
    rank, ws = dist.get_rank(), dist.get_world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    from vescale_amd.dmodule.ring_attention import ring_sdpa
    from vescale_amd.dmodule.ulysses import ulysses_sdpa
    from vescale_amd.dtensor import DTensor, Shard, init_device_mesh

    mesh = init_device_mesh(device.type, (ws,), mesh_dim_names=("SP",))
    sp_fn = ulysses_sdpa if args.algo == "ulysses" else ring_sdpa

    def attn(q, k, v):
        # locals [B,H,S/p,D] -> DTensor Shard(2) -> exact SP attention
        dq = DTensor.from_local(q, mesh, [Shard(2)])
        dk = DTensor.from_local(k, mesh, [Shard(2)])
        dv = DTensor.from_local(v, mesh, [Shard(2)])
        return sp_fn(dq, dk, dv, is_causal=True).to_local()

    torch.manual_seed(11)
    model = SPBlock(args.dim, args.heads, attn).to(device)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    g = torch.Generator().manual_seed(1234)
    s_local = args.seq // ws
    assert args.seq % ws == 0 and args.heads % ws == 0
    for step in range(args.steps):
        x = torch.randn(2, args.seq, args.dim, generator=g)  # full seq, all ranks
        xl = x.narrow(1, rank * s_local, s_local).to(device)  # my shard
        out = model(xl)
        loss = out.pow(2).mean()  # local mean; allreduce-avg = global mean
        opt.zero_grad(set_to_none=True)
        loss.backward()
        # DP-free run: average the replicated params' grads over the seq shards
        for p in model.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad)
                p.grad /= ws
        opt.step()
        lt = loss.detach().clone()
        dist.all_reduce(lt)
        if rank == 0:
            print(f"[{args.algo}] step {step}: loss {float(lt) / ws:.6f}", flush=True)
    dist.barrier()
    if rank == 0:
        print("DONE")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
