"""Gradient clipping over flat/sharded grads.

Parity: legacy/vescale/optim/clip_grads.py:21 clip_grad_norm_fp32 —
local L2^2 via the fused CDNA4 kernel, one allreduce over the given
groups, in-place scale.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from ..ops import l2norm_sq, scale_flat_


@torch.no_grad()
def clip_grad_norm_fp32(
    grads: List[torch.Tensor],
    max_norm: float,
    pgs: Optional[List] = None,
) -> torch.Tensor:
    """grads: LOCAL flat tensors (each rank's shard).  pgs: process groups
    over which the norm must be summed (DP shard group, TP group...).
    Returns the global grad norm (fp32 scalar tensor)."""
    if not grads:
        total = torch.zeros((), dtype=torch.float32)
    else:
        total = None
        for g in grads:
            s = l2norm_sq(g)
            total = s if total is None else total + s
    for pg in pgs or []:
        if pg is not None and dist.is_initialized():
            dist.all_reduce(total, group=pg)
    norm = total.sqrt()
    scale = (max_norm / (norm + 1.0e-6)).clamp(max=1.0)
    for g in grads:
        scale_flat_(g, scale)
    return norm
