"""Pinned-memory D2H pool for async checkpointing.

Parity: legacy/vescale/checkpoint/utilities/mem_checkpoint.py:66-140
(PinnedStoragePool / copy_gpu_tensor_to_cpu_pinned_mem_pool): device
tensors are staged through reusable pinned host buffers so the D2H copy
is async w.r.t. compute and the expensive pinned allocation is amortized
across checkpoints.
"""
from __future__ import annotations

import threading
from typing import Dict, List

import torch


class PinnedStoragePool:
    def __init__(self):
        self._lock = threading.Lock()
        self._free: Dict[int, List[torch.Tensor]] = {}

    def acquire(self, nbytes: int) -> torch.Tensor:
        size = 1 << max(12, (nbytes - 1).bit_length())  # round to pow2 >= 4k
        with self._lock:
            lst = self._free.get(size)
            if lst:
                return lst.pop()
        try:
            return torch.empty(size, dtype=torch.uint8, pin_memory=True)
        except RuntimeError:
            return torch.empty(size, dtype=torch.uint8)

    def release(self, buf: torch.Tensor):
        with self._lock:
            self._free.setdefault(buf.numel(), []).append(buf)

    def clear(self):
        with self._lock:
            self._free.clear()


GLOBAL_POOL = PinnedStoragePool()


def copy_gpu_tensor_to_cpu_pinned_mem_pool(t: torch.Tensor, non_blocking: bool = True) -> torch.Tensor:
    """Stage a device tensor into a pooled pinned buffer; returns a CPU
    tensor viewing that buffer (caller releases via release_cpu_tensor)."""
    if not t.is_cuda:
        return t.detach().clone()
    nbytes = t.numel() * t.element_size()
    buf = GLOBAL_POOL.acquire(nbytes)
    flat = buf.narrow(0, 0, nbytes).view(t.dtype).reshape(t.shape) if nbytes else buf[:0].view(t.dtype)
    flat.copy_(t.detach(), non_blocking=non_blocking)
    flat._pool_buf = buf  # keep the backing buffer alive
    return flat


def release_cpu_tensor(t: torch.Tensor):
    buf = getattr(t, "_pool_buf", None)
    if buf is not None:
        GLOBAL_POOL.release(buf)


def mem_checkpoint_path(name: str = "vescale_amd_ckpt") -> str:
    """An IN-MEMORY checkpoint directory (tmpfs).

    Parity: the reference ships a gRPC in-memory file server
    (checkpoint/utilities/server/mem_server_lib.py) so checkpoints can be
    staged to RAM.  On the MI355X single-node deployment the same
    capability is the host's tmpfs: DCP save/load against /dev/shm is an
    in-memory checkpoint with zero extra machinery (cross-host restore
    still goes through the shared filesystem path).
    """
    import os

    base = "/dev/shm" if os.path.isdir("/dev/shm") else None
    if base is None:
        import tempfile

        base = tempfile.gettempdir()
    path = os.path.join(base, name)
    os.makedirs(path, exist_ok=True)
    return path
