"""Mixtral expert-parallel training example (parity role:
legacy/examples/mixtral_EP_training/).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
  --master-addr 127.0.0.1 examples/mixtral_ep_train.py
"""
import argparse
import os

import sys

sys.path.insert(0, __file__.rsplit("/examples/", 1)[0])

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--rebalance-every", type=int, default=0,
                    help="dynamically re-place experts by observed load "
                         "every N steps (moe.rebalance_experts)")
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    dist.init_process_group(os.environ.get("VESCALE_BACKEND", "nccl" if on_gpu else "gloo"))
    rank = dist.get_rank()
    world = dist.get_world_size()
    device = torch.device("cuda") if on_gpu else torch.device("cpu")

    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.models.mixtral import MixtralModel, mixtral_8x7b, mixtral_tiny
    from vescale_amd.moe import (
        BasicExpertsAllocator,
        LoadBalancedExpertsAllocator,
        parallelize_experts,
    )

    mesh = init_device_mesh(device.type, (world,), mesh_dim_names=("EP",))
    torch.manual_seed(0)
    cfg = mixtral_8x7b() if on_gpu else mixtral_tiny()
    if not on_gpu:
        args.seq = 32
    model = MixtralModel(cfg).to(device)
    model.init_weights()
    parallelize_experts(
        model, mesh,
        allocator_cls=(LoadBalancedExpertsAllocator if args.rebalance_every
                       else BasicExpertsAllocator),
    )
    # non-expert params are replicated -> plain DDP-style allreduce of their
    # grads; expert grads already carry all EP ranks' token contributions
    opt = torch.optim.AdamW([p for p in model.parameters()], lr=3e-4)

    gen = torch.Generator().manual_seed(rank)
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), generator=gen).to(device)
        y = torch.roll(x, -1, dims=1)
        loss = model(x, y)
        loss.backward()
        # sync non-expert grads across EP (they act as DP for dense params)
        for n, p in model.named_parameters():
            if p.grad is not None and ".experts." not in n:
                dist.all_reduce(p.grad, group=mesh.get_group(0))
                p.grad /= world
        opt.step()
        opt.zero_grad()
        # dynamic expert re-placement from observed routing load
        if args.rebalance_every and (step + 1) % args.rebalance_every == 0:
            from vescale_amd.moe import rebalance_experts

            for layer in model.layers:
                moe = layer.moe
                with torch.no_grad():
                    logits = moe.router(
                        model.tok_embeddings(x).reshape(-1, cfg.dim)
                    )
                    ids = torch.topk(logits, cfg.top_k, dim=-1).indices.reshape(-1)
                    counts = torch.bincount(ids.cpu(), minlength=cfg.n_experts)
                moved = rebalance_experts(moe, counts.tolist(), optimizer=opt)
                if moved and rank == 0:
                    print(f"step {step}: re-placed experts "
                          f"{moe.allocator.placement}", flush=True)
        if rank == 0 and step % 5 == 0:
            print(f"step {step} loss {float(loss.detach()):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
