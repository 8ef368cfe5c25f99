"""Emulated mesh collectives mirroring dtensor/_collective_utils
(parity: legacy/vescale/emulator/mesh_collectives.py:24-178)."""
from __future__ import annotations

from typing import List

import torch

from .distributed import EmulatorProcessGroup


def emu_all_gather(pg: EmulatorProcessGroup, tensors: List[torch.Tensor], dim: int = 0):
    full = torch.cat(tensors, dim=dim)
    return [full.clone() for _ in range(pg.size())]


def emu_reduce_scatter(pg: EmulatorProcessGroup, tensors: List[torch.Tensor], dim: int = 0):
    W = pg.size()
    red = tensors[0].clone()
    for t in tensors[1:]:
        red = red + t
    return [c.contiguous() for c in torch.chunk(red, W, dim=dim)]


def emu_all_to_all(pg: EmulatorProcessGroup, chunk_lists: List[List[torch.Tensor]]):
    return pg.all_to_all(chunk_lists)
