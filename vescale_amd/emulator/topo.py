"""Collective topologies (parity: legacy/vescale/emulator/topo.py:83 —
ring order + double binary trees, the two RCCL/NCCL algorithms).

xGMI note: MI355X intra-node is FULLY CONNECTED (7 p2p links/GPU), so the
"ring" is any Hamiltonian cycle — RCCL uses multiple rings (one per
link); the emulator models a single logical ring (ordering-equivalent
for bitwise verification) plus the double binary tree and the one-shot
direct algorithm RCCL prefers for small messages on all-pairs fabrics.
"""
from __future__ import annotations

from typing import List, Optional, Tuple


def ring_order(world: int) -> List[int]:
    return list(range(world))


def _tree_parent(rank: int, world: int) -> Optional[int]:
    """Binary tree 0: standard NCCL up-tree built on bit tricks."""
    if rank == 0:
        return None
    # lowest set bit
    low = rank & (-rank)
    parent = rank ^ low
    if parent == rank:
        return None
    return parent


def double_binary_trees(world: int) -> Tuple[dict, dict]:
    """Two complementary trees (parity: NCCL double binary tree).  Tree A
    from bit structure of rank; tree B from shifted ranks so every rank is
    a leaf in one tree."""
    tree_a = {r: _tree_parent(r, world) for r in range(world)}
    shift = 1 if world > 1 else 0
    tree_b = {
        r: (
            None
            if ((r + shift) % world) == 0
            else ((_tree_parent((r + shift) % world, world) - shift) % world
                  if _tree_parent((r + shift) % world, world) is not None
                  else None)
        )
        for r in range(world)
    }
    return tree_a, tree_b


def tree_children(parents: dict) -> dict:
    out = {r: [] for r in parents}
    for r, p in parents.items():
        if p is not None:
            out[p].append(r)
    return out
