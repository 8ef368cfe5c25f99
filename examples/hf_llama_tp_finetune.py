"""Fine-tune an UNMODIFIED HuggingFace Llama with tensor parallelism.

Parity: legacy/examples/llama2_4D_finetune/llama_train.py — the reference
parallelizes a stock transformers model purely through sharding plans.
Same here, against transformers >= 5: no model edits, one plan.

Run (CPU smoke, 2 ranks):
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
        examples/hf_llama_tp_finetune.py --steps 5
On MI355X replace nothing: the same script runs per-GPU ranks over RCCL.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--seq", type=int, default=64)
    ap.add_argument("--lr", type=float, default=1e-4)
    args = ap.parse_args()

    if not dist.is_initialized():
        backend = os.environ.get("VESCALE_BACKEND", "nccl" if torch.cuda.is_available() else "gloo")
        dist.init_process_group(backend)
    rank = dist.get_rank()
    ws = dist.get_world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    from transformers import LlamaConfig, LlamaForCausalLM

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.models.hf_llama_plan import hf_llama_tp_plan

    cfg = LlamaConfig(
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=4,
        num_attention_heads=8,
        num_key_value_heads=4,
        vocab_size=1024,
        attn_implementation="eager",
    )
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg).to(device)
    mesh = init_device_mesh(device.type, (ws,), mesh_dim_names=("tp",))
    model = parallelize_module(model, mesh, hf_llama_tp_plan())

    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)
    g = torch.Generator().manual_seed(1234)
    for step in range(args.steps):
        ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), generator=g).to(device)
        out = model(input_ids=ids, labels=ids)
        loss = out.loss
        with torch.no_grad():
            loss_val = float(loss.full_tensor() if hasattr(loss, "full_tensor") else loss)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        if rank == 0:
            print(f"step {step}: loss {loss_val:.4f}", flush=True)
    dist.barrier()
    if rank == 0:
        print("DONE")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
