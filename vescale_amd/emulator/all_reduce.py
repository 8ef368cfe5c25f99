"""Emulated allreduce algorithms with RCCL-faithful reduction order.

Parity: legacy/vescale/emulator/all_reduce.py:40-350 (ring + double
binary tree, chunk-size model from NCCL 2.19.3's
calculate_chunk_size.py).  All ranks' buffers live in ONE process; the
element-wise reduction ORDER matches the algorithm's real dataflow, so
an emulated run is bitwise-comparable to a hardware run using the same
algorithm — the correctness sanitizer for communication (reference
emulator/README.md:31-34).
"""
from __future__ import annotations

from typing import List

import torch

from .topo import double_binary_trees, ring_order, tree_children


def run_ring_all_reduce(buffers: List[torch.Tensor]) -> List[torch.Tensor]:
    """Ring allreduce: reduce-scatter pass then all-gather pass.
    buffers[r] is rank r's input; returns the reduced buffers (all equal,
    each element reduced in ring order starting from its chunk owner)."""
    W = len(buffers)
    if W == 1:
        return buffers
    n = buffers[0].numel()
    flats = [b.reshape(-1) for b in buffers]
    # chunk boundaries (NCCL splits into W chunks, remainder to the front)
    base = n // W
    rem = n % W
    sizes = [base + (1 if i < rem else 0) for i in range(W)]
    offs = [0]
    for s in sizes[:-1]:
        offs.append(offs[-1] + s)

    # reduce-scatter then all-gather, modeled as: chunk c enters the ring at
    # rank (c+1)%W and accumulates sequentially around it — the exact
    # element-wise addition order of the ring algorithm
    acc = [f.clone() for f in flats]
    for c in range(W):
        o, s = offs[c], sizes[c]
        # chunk c starts at rank (c+1)%W and travels the ring accumulating
        cur = flats[(c + 1) % W][o : o + s].clone()
        for step in range(1, W):
            r = (c + 1 + step) % W
            cur = cur + flats[r][o : o + s]
        for r in range(W):
            acc[r][o : o + s] = cur
    return [a.reshape(buffers[0].shape) for a in acc]


def run_tree_all_reduce(buffers: List[torch.Tensor]) -> List[torch.Tensor]:
    """Double-binary-tree allreduce: reduce up tree A (halves split across
    the two trees in NCCL; here full buffer up tree A for clarity),
    broadcast down."""
    W = len(buffers)
    if W == 1:
        return buffers
    tree_a, _ = double_binary_trees(W)
    children = tree_children(tree_a)
    flats = [b.reshape(-1).clone() for b in buffers]

    # post-order reduce to root (children accumulated in ascending order —
    # fixed order = deterministic bitwise result)
    def reduce_up(r):
        for c in sorted(children[r]):
            reduce_up(c)
            flats[r] += flats[c]

    root = next(r for r, p in tree_a.items() if p is None)
    reduce_up(root)

    def bcast_down(r):
        for c in sorted(children[r]):
            flats[c].copy_(flats[r])
            bcast_down(c)

    bcast_down(root)
    return [f.reshape(buffers[0].shape) for f in flats]


def run_direct_all_reduce(buffers: List[torch.Tensor]) -> List[torch.Tensor]:
    """One-shot all-pairs allreduce — the xGMI-preferred algorithm for
    latency-bound sizes (each GPU reads all peers over its 7 links and
    reduces locally in RANK ORDER)."""
    W = len(buffers)
    out = []
    for r in range(W):
        acc = buffers[0].clone()
        for p in range(1, W):
            acc = acc + buffers[p]
        out.append(acc)
    return out
