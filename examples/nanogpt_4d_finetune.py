"""nanoGPT 4D (DP x TP/SP + ZeRO2) training example.

Parity role: legacy/examples/nanogpt_4D_finetune/finetune_4D.py — the
same wiring: VeDeviceMesh -> parallelize_module(TP) -> DDP(DP) ->
DistributedOptimizer, loss-parity methodology vs 1 GPU.

Run (CPU proof, 4 procs): python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 4 --master-addr 127.0.0.1 examples/nanogpt_4d_finetune.py
"""
import argparse
import os

import sys

sys.path.insert(0, __file__.rsplit("/examples/", 1)[0])

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dp", type=int, default=2)
    ap.add_argument("--tp", type=int, default=2)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--batch", type=int, default=8)
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    dist.init_process_group(os.environ.get("VESCALE_BACKEND", "nccl" if on_gpu else "gloo"))
    rank = dist.get_rank()
    device = "cuda" if on_gpu else "cpu"

    from vescale_amd.devicemesh_api import VESCALE_DEVICE_MESH as VMESH
    from vescale_amd.ddp import DistributedDataParallel as DDP
    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import DTensor, Replicate
    from vescale_amd.models.nanogpt import GPT, gpt_tiny, gpt2_small
    from vescale_amd.models.nanogpt_plan import nanogpt_tp_plan
    from vescale_amd.optim import DistributedOptimizer

    mesh = VMESH.init_device_mesh(device, (args.dp, args.tp), mesh_dim_names=("DP", "TP"))
    torch.manual_seed(0)
    cfg = gpt2_small() if on_gpu else gpt_tiny()
    model = GPT(cfg).to(device)
    parallelize_module(model, mesh["TP"], nanogpt_tp_plan(sp=True))
    ddp = DDP(model, mesh.get_group(0), use_distributed_optimizer=True)
    opt = DistributedOptimizer(
        torch.optim.AdamW(model.parameters(), lr=3e-4), [ddp], clip_grad=1.0,
        extra_norm_pgs=[mesh.get_group(1)],
    )

    dp_rank = VMESH.get_data_parallel_rank()
    gen = torch.Generator().manual_seed(dp_rank)
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch // args.dp, 32), generator=gen).to(device)
        y = torch.randint(0, cfg.vocab_size, (args.batch // args.dp, 32), generator=gen).to(device)
        _, loss = ddp(x, y)
        loss.backward()
        model.finish_grad_sync()
        opt.step()
        opt.zero_grad()
        lv = loss
        if isinstance(lv, DTensor):
            lv = lv.redistribute(placements=[Replicate()]).to_local()
        if rank == 0 and step % 5 == 0:
            print(f"step {step} loss {float(lv):.4f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
