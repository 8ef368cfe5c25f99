"""Global-view emulated process group — all ranks in one process.

Parity: legacy/vescale/emulator/distributed.py:52-700 (ProcessGroup,
_World, init_process_group) + emulator_instrumentation.py:81 (rewrites
torch.distributed calls to loop over ranks).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from .all_reduce import run_direct_all_reduce, run_ring_all_reduce, run_tree_all_reduce


class EmulatorProcessGroup:
    """Holds every rank's tensors; collective methods take LISTS indexed by
    rank and mutate/return per-rank results."""

    def __init__(self, world_size: int, algo: str = "ring"):
        self.world_size = world_size
        self.algo = algo

    def size(self) -> int:
        return self.world_size

    # ------------------------------------------------------------------
    def all_reduce(self, tensors: List[torch.Tensor], op: str = "sum"):
        assert len(tensors) == self.world_size
        assert op == "sum", "emulator models sum reductions"
        if self.algo == "ring":
            out = run_ring_all_reduce(tensors)
        elif self.algo == "tree":
            out = run_tree_all_reduce(tensors)
        else:
            out = run_direct_all_reduce(tensors)
        for t, o in zip(tensors, out):
            t.copy_(o)
        return tensors

    def all_gather(self, tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        full = torch.cat([t.reshape(-1) for t in tensors])
        return [full.clone() for _ in range(self.world_size)]

    def reduce_scatter(self, tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        W = self.world_size
        reduced = tensors[0].clone()
        for t in tensors[1:]:
            reduced = reduced + t
        chunks = reduced.reshape(W, -1)
        return [chunks[r].clone() for r in range(W)]

    def all_to_all(self, per_rank_chunks: List[List[torch.Tensor]]) -> List[List[torch.Tensor]]:
        W = self.world_size
        return [[per_rank_chunks[src][dst] for src in range(W)] for dst in range(W)]

    def broadcast(self, tensors: List[torch.Tensor], src: int) -> List[torch.Tensor]:
        for r in range(self.world_size):
            if r != src:
                tensors[r].copy_(tensors[src])
        return tensors


_world: Optional[EmulatorProcessGroup] = None


def init_emulator(world_size: int, algo: str = "ring") -> EmulatorProcessGroup:
    global _world
    _world = EmulatorProcessGroup(world_size, algo)
    return _world


def get_emulator() -> Optional[EmulatorProcessGroup]:
    return _world
