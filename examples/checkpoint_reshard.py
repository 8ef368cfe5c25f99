"""Cross-topology checkpoint example: train Llama-tiny with FSDP, save,
then reload the SAME checkpoint (model + per-param optimizer state) at a
DIFFERENT world size — the reference's load-time resharding capability
(checkpoint/README features) on the MI355X stack.

Run (phase 1, 4 ranks):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 examples/checkpoint_reshard.py --save /tmp/ck
Run (phase 2, 2 ranks — NOTE the different world size):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 examples/checkpoint_reshard.py --load /tmp/ck
"""
import argparse
import os
import sys

sys.path.insert(0, __file__.rsplit("/examples/", 1)[0])

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--save", type=str, default=None)
    ap.add_argument("--load", type=str, default=None)
    ap.add_argument("--steps", type=int, default=5)
    args = ap.parse_args()
    assert bool(args.save) != bool(args.load), "pass exactly one of --save/--load"

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    on_gpu = torch.cuda.is_available()
    dist.init_process_group(os.environ.get("VESCALE_BACKEND", "nccl" if on_gpu else "gloo"))
    rank = dist.get_rank()
    ws = dist.get_world_size()
    dev = torch.device("cuda", rank % max(torch.cuda.device_count(), 1)) if on_gpu else torch.device("cpu")
    if on_gpu:
        torch.cuda.set_device(dev)

    import vescale_amd.checkpoint as ckpt
    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.fsdp import FSDP, FlatAdamW
    from vescale_amd.models.llama import LlamaModel, llama_tiny

    torch.manual_seed(7)
    cfg = llama_tiny()
    model = LlamaModel(cfg)
    model.init_weights()
    mesh = init_device_mesh(dev.type if dev.type == "cuda" else "cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.bfloat16 if on_gpu else torch.float32, device=dev)
    opt = FlatAdamW(eng, lr=1e-3, grad_clip=1.0)

    g = torch.Generator().manual_seed(99)
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (2, 64), generator=g).to(dev)
        loss = eng(x, torch.roll(x, -1, 1))
        loss.backward()
        opt.step()
        if rank == 0 and step % 2 == 0:
            print(f"[ws{ws}] step {step} loss {float(loss.detach()):.4f}", flush=True)

    if args.save:
        ckpt.save(args.save, {"model": eng, "optimizer": opt})
        if rank == 0:
            print(f"saved at world size {ws} -> {args.save}", flush=True)
    else:
        ckpt.load(args.load, {"model": eng, "optimizer": opt})
        if rank == 0:
            print(f"loaded a checkpoint into world size {ws} (resharded)", flush=True)
        # continue training to show the state is live
        x = torch.randint(0, cfg.vocab_size, (2, 64), generator=g).to(dev)
        loss = eng(x, torch.roll(x, -1, 1))
        loss.backward()
        opt.step()
        if rank == 0:
            print(f"post-load step loss {float(loss.detach()):.4f}", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
