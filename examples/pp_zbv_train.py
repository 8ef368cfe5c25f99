"""Pipeline-parallel training with the ZB-V zero-bubble schedule.

Run (CPU smoke, 4 stages x 2 virtual chunks in a V placement):
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 4 \
        examples/pp_zbv_train.py --schedule zero_bubble_v --steps 5
Other schedules: 1f1b, gpipe, interleaved_1f1b (vc=2).
On MI355X the same script runs one rank per GPU over RCCL.

The model is a stack of MLP blocks split uniformly over
num_stages x virtual_chunks parts; ZB-V places chunk 0 down the ranks and
chunk 1 back up (rank 0 computes the loss), with the Linear weight-grad
GEMMs deferred into bubble-filling W phases (pipe/wgrad_store.py).
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn as nn


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--schedule", default="zero_bubble_v",
                    choices=["1f1b", "gpipe", "interleaved_1f1b", "zero_bubble_v"])
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--microbatches", type=int, default=4)
    ap.add_argument("--dim", type=int, default=64)
    ap.add_argument("--blocks", type=int, default=8)
    args = ap.parse_args()

    backend = os.environ.get("VESCALE_BACKEND", "nccl" if torch.cuda.is_available() else "gloo")
    dist.init_process_group(backend)
    rank, ws = dist.get_rank(), dist.get_world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage
    from vescale_amd.plan import (
        PipelineParallelPlan,
        PipelineScheduleType,
        PipelineSplitMethodType,
    )

    vc = 2 if args.schedule in ("interleaved_1f1b", "zero_bubble_v") else 1
    torch.manual_seed(17)
    blocks = [
        nn.Sequential(nn.Linear(args.dim, args.dim), nn.Tanh())
        for _ in range(args.blocks)
    ]
    plan = PipelineParallelPlan(
        num_stages=ws,
        virtual_chunks=vc,
        schedule_type=PipelineScheduleType(args.schedule),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(blocks, plan, rank).to(device)
    loss_fn = lambda out, tgt: (out - tgt).pow(2).mean()  # noqa: E731
    engine = PipeEngine(stage, plan, loss_fn=loss_fn, device=device)
    opt = torch.optim.AdamW(stage.parameters(), lr=1e-3)

    g = torch.Generator().manual_seed(1234)
    for step in range(args.steps):
        x = torch.randn(8, args.dim, generator=g)
        y = torch.randn(8, args.dim, generator=g)
        opt.zero_grad(set_to_none=True)
        loss = engine.forward_backward((x, y), args.microbatches)
        opt.step()
        # ZB-V computes the loss on rank 0; 1F1B/GPipe on the last rank
        if loss is not None:
            print(f"[rank {rank}] step {step}: loss {float(loss):.5f}", flush=True)
    dist.barrier()
    if rank == 0:
        print("DONE")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
