import os, sys, torch
import torch.distributed as dist
import torch.multiprocessing as mp

def run(rank, ws, port):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank), WORLD_SIZE=str(ws))
    try:
        dist.init_process_group("nccl", rank=rank, world_size=ws)
        torch.cuda.set_device(0)
        t = torch.ones(4, device="cuda") * (rank + 1)
        dist.all_reduce(t)
        print(f"[r{rank}] allreduce -> {t.tolist()}", flush=True)
        dist.barrier()
        dist.destroy_process_group()
        print(f"[r{rank}] OK", flush=True)
    except Exception as e:
        print(f"[r{rank}] FAIL: {type(e).__name__}: {str(e)[:300]}", flush=True)

if __name__ == "__main__":
    ctx = mp.get_context("spawn")
    ps = [ctx.Process(target=run, args=(r, 2, 29721)) for r in range(2)]
    [p.start() for p in ps]
    [p.join(90) for p in ps]
    [p.terminate() for p in ps if p.is_alive()]
