"""Multi-process test harness (parity concept: legacy/test/common_dtensor.py
DTensorTestBase/with_comms — redesigned as a light spawn helper).

CPU tests run world_size>1 over gloo with fork-start processes (fast, no
CUDA in parent).  GPU tests use spawn + nccl(RCCL).
"""
from __future__ import annotations

import os
import socket
import traceback
from typing import Callable

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _entry(rank: int, world_size: int, port: int, backend: str, fn, args, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_RANK"] = str(rank)
        dist.init_process_group(backend, rank=rank, world_size=world_size)
        torch.manual_seed(0)
        fn(rank, world_size, *args)
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, None))
    except Exception:
        q.put((rank, traceback.format_exc()))
        raise


def spawn(world_size: int, fn: Callable, *args, backend: str = "gloo"):
    """Run fn(rank, world_size, *args) on world_size processes."""
    port = _free_port()
    ctx = mp.get_context("fork" if backend == "gloo" else "spawn")
    q = ctx.SimpleQueue()
    procs = []
    for r in range(world_size):
        p = ctx.Process(target=_entry, args=(r, world_size, port, backend, fn, args, q))
        p.start()
        procs.append(p)
    errs = []
    for _ in range(world_size):
        rank, err = q.get()
        if err is not None:
            errs.append(f"--- rank {rank} ---\n{err}")
    for p in procs:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
            errs.append(f"rank proc {p.pid} hung")
    assert not errs, "\n".join(errs)
