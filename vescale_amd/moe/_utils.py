"""MoE comm utilities.

Parity: legacy/vescale/moe/_utils.py:26-59 — global_all_to_all_single,
the autograd-wrapped token shuffle.  On xGMI the all_to_all is the
natural fit: tokens fan out over all 7 p2p links concurrently.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.distributed as dist


class _AllToAllSingle(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        out_n = sum(out_splits) if out_splits is not None else x.shape[0]
        out = x.new_empty((out_n,) + tuple(x.shape[1:]))
        if group is None or dist.get_world_size(group) == 1:
            out.copy_(x)
            return out
        try:
            dist.all_to_all_single(
                out, x.contiguous(),
                list(out_splits) if out_splits else None,
                list(in_splits) if in_splits else None,
                group=group,
            )
        except RuntimeError:
            _a2a_fallback(out, x.contiguous(), out_splits, in_splits, group)
        return out

    @staticmethod
    def backward(ctx, g):
        back = _AllToAllSingle.apply(g, ctx.in_splits, ctx.out_splits, ctx.group)
        return back, None, None, None


def _a2a_fallback(out, inp, out_splits, in_splits, group):
    """gloo: pairwise isend/irecv.  CUDA tensors are staged through CPU —
    gloo point-to-point on device memory is outside its supported envelope
    (observed to take down a node when two CUDA ranks shared a GPU)."""
    if inp.is_cuda:
        cpu_out = torch.empty(out.shape, dtype=out.dtype)
        _a2a_fallback(cpu_out, inp.cpu(), out_splits, in_splits, group)
        out.copy_(cpu_out)
        return
    ws = dist.get_world_size(group)
    me = dist.get_rank(group)
    iss = list(in_splits) if in_splits else [inp.shape[0] // ws] * ws
    oss = list(out_splits) if out_splits else [out.shape[0] // ws] * ws
    in_chunks, off = [], 0
    for s in iss:
        in_chunks.append(inp.narrow(0, off, s))
        off += s
    out_chunks, off = [], 0
    for s in oss:
        out_chunks.append(out.narrow(0, off, s))
        off += s
    out_chunks[me].copy_(in_chunks[me])
    reqs = []
    for peer in range(ws):
        if peer == me:
            continue
        gr = dist.get_global_rank(group, peer)
        if out_chunks[peer].numel():
            reqs.append(dist.irecv(out_chunks[peer].contiguous() if not out_chunks[peer].is_contiguous() else out_chunks[peer], src=gr, group=group))
        if in_chunks[peer].numel():
            reqs.append(dist.isend(in_chunks[peer].contiguous(), dst=gr, group=group))
    for r in reqs:
        r.wait()


def global_all_to_all_single(
    x: torch.Tensor,
    output_split_sizes: Optional[Sequence[int]] = None,
    input_split_sizes: Optional[Sequence[int]] = None,
    group=None,
) -> torch.Tensor:
    """Differentiable all_to_all_single over dim 0 (the token shuffle)."""
    return _AllToAllSingle.apply(
        x,
        list(output_split_sizes) if output_split_sizes is not None else None,
        list(input_split_sizes) if input_split_sizes is not None else None,
        group,
    )
