"""Llama-3 FSDP training example (the flagship path; parity role:
reference open_llama_4D_benchmark / llama2_4D_finetune).

1 GPU:  python examples/llama3_fsdp_train.py --model llama_tiny
8 GPU:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
          --master-addr 127.0.0.1 examples/llama3_fsdp_train.py
"""
import argparse
import os
import time

import sys

sys.path.insert(0, __file__.rsplit("/examples/", 1)[0])

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3_8b")
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--seq", type=int, default=8192)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--ckpt-dir", default=None)
    ap.add_argument("--ndtimeline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{os.environ.get('LOCAL_RANK', 0)}") if on_gpu else torch.device("cpu")
    if world > 1:
        dist.init_process_group(os.environ.get("VESCALE_BACKEND", "nccl" if on_gpu else "gloo"))

    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.fsdp import FSDP, FlatAdamW
    from vescale_amd.models import llama as M
    import vescale_amd.checkpoint as ckpt

    if args.ndtimeline:
        from vescale_amd.ndtimeline import init_ndtimers

        mgr = init_ndtimers(chrome_trace_path="ndtimeline_trace.json",
                            summary=True)

    cfg = getattr(M, args.model)() if args.model != "llama_tiny" else M.llama_tiny()
    if not on_gpu:
        cfg = M.llama_tiny()
        args.batch, args.seq = 2, 64

    mesh = init_device_mesh(device.type, (world,), mesh_dim_names=("DP",)) if world > 1 else None
    torch.manual_seed(1234)
    model = M.LlamaModel(cfg).to(device)
    model.init_weights()
    eng = FSDP(model, mesh, param_dtype=torch.bfloat16 if on_gpu else torch.float32, device=device)
    opt = FlatAdamW(eng, lr=args.lr, grad_clip=1.0)

    gen = torch.Generator().manual_seed(rank)
    t0 = time.perf_counter()
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), generator=gen).to(device)
        y = torch.roll(x, -1, dims=1)
        loss = eng(x, y)
        loss.backward()
        opt.step()
        if rank == 0 and step % 10 == 0:
            el = time.perf_counter() - t0
            toks = args.batch * args.seq * world * (step + 1)
            print(f"step {step} loss {float(loss.detach()):.4f} tok/s {toks/el:,.0f}")
    if args.ckpt_dir:
        ckpt.save(args.ckpt_dir, {"model": eng})
        if rank == 0:
            print("checkpoint saved to", args.ckpt_dir)
    if args.ndtimeline:
        from vescale_amd.ndtimeline import flush, wait

        flush(args.steps)
        wait()
        sh = getattr(mgr, "summary_handler", None)
        if sh is not None and rank == 0:
            import json as _json

            top = sorted(sh.summary().items(),
                         key=lambda kv: -kv[1]["total_us"])[:8]
            print("ndtimeline summary (top by total time):")
            for k, v in top:
                print(f"  {k:<28} n={v['count']:<5} mean={v['mean_us']:.0f}us "
                      f"p99={v['p99_us']:.0f}us total={v['total_us']/1e3:.1f}ms")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
