"""parallelize_experts — expert-parallelize a model's MoE layers.

Parity: legacy/vescale/moe/api.py:39 + _experts.py + _scheduler.py — each
EP rank keeps only the experts the allocator assigns it; the token
shuffle is one uneven all_to_all each way (xGMI all-pairs traffic), with
per-(rank, expert) counts exchanged on a small int tensor first.
"""
from __future__ import annotations

import re
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..dtensor import DeviceMesh
from .experts_allocator import BasicExpertsAllocator, ExpertsAllocator
from .token_dispatcher import BasicTokenDispatcher, TokenDispatcher


def parallelize_experts(
    model: nn.Module,
    ep_mesh_or_group,
    *,
    layer_cls: str = "MoELayer",
    allocator_cls=BasicExpertsAllocator,
    dispatcher_cls=BasicTokenDispatcher,
) -> nn.Module:
    if isinstance(ep_mesh_or_group, DeviceMesh):
        group = ep_mesh_or_group.get_group(ep_mesh_or_group.ndim - 1)
    else:
        group = ep_mesh_or_group
    ws = dist.get_world_size(group) if (group is not None and dist.is_initialized()) else 1
    rank = dist.get_rank(group) if (group is not None and dist.is_initialized()) else 0

    for name, mod in model.named_modules():
        if type(mod).__name__ == layer_cls:
            n_exp = len(mod.experts)
            allocator = allocator_cls(n_exp, ws)
            dispatcher = dispatcher_cls(allocator)
            local_ids = allocator.experts_of(rank)
            # prune remote experts (their params are freed on this rank)
            for e in range(n_exp):
                if e not in local_ids:
                    mod.experts[e] = _RemoteExpert()
            mod.enable_expert_parallel(group, allocator, dispatcher, local_ids)
    return model


class _RemoteExpert(nn.Module):
    """Placeholder for an expert owned by another EP rank."""

    def forward(self, x):  # pragma: no cover
        raise RuntimeError("expert not resident on this rank")
