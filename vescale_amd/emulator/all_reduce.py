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


def run_ring_all_reduce(
    buffers: List[torch.Tensor], chunk_bytes: int = 0,
    order: "List[int]" = None,
) -> List[torch.Tensor]:
    """Ring allreduce: reduce-scatter pass then all-gather pass.
    buffers[r] is rank r's input; returns the reduced buffers (all equal,
    each element reduced in ring order starting from its chunk owner).

    `order` (from a parsed NCCL_GRAPH_DUMP_FILE ring channel,
    graph_dump.TopoGraph.ring()) replaces the identity ring with the
    machine's actual ring order — the reduction order then matches the
    hardware's bitwise.

    chunk_bytes > 0 enables the CHUNKED geometry the real collective uses
    (emulator/calculate_chunk_size.py): the buffer is processed in loops
    of nranks * chunk elements and each element's reduction order follows
    its position WITHIN ITS LOOP — which is what decides the floating
    point addition order on hardware for buffers larger than one loop."""
    W = len(buffers)
    if W == 1:
        return buffers
    n = buffers[0].numel()
    flats = [b.reshape(-1) for b in buffers]
    acc = [f.clone() for f in flats]

    def loop_segments():
        if chunk_bytes > 0:
            eb = buffers[0].element_size()
            chunk = max(1, chunk_bytes // eb)
            loop_elems = W * chunk
            loops = []
            off = 0
            while off < n:
                this_loop = min(loop_elems, n - off)
                base = this_loop // W
                rem = this_loop % W
                segs = []
                o = off
                for c in range(W):
                    s = base + (1 if c < rem else 0)
                    segs.append((o, s))
                    o += s
                loops.append(segs)
                off += this_loop
            return loops
        # unchunked: one loop spanning the buffer, W chunks
        base = n // W
        rem = n % W
        segs = []
        o = 0
        for c in range(W):
            s = base + (1 if c < rem else 0)
            segs.append((o, s))
            o += s
        return [segs]

    ring = list(order) if order is not None else ring_order(W)
    assert sorted(ring) == list(range(W)), f"bad ring order {ring}"
    for segs in loop_segments():
        for c, (o, s) in enumerate(segs):
            if s == 0:
                continue
            # chunk c starts at ring position c+1 and travels the ring
            cur = flats[ring[(c + 1) % W]][o : o + s].clone()
            for step in range(1, W):
                r = ring[(c + 1 + step) % W]
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
