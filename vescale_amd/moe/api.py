"""parallelize_experts — expert-parallelize a model's MoE layers.

Parity: legacy/vescale/moe/api.py:39 + _experts.py + _scheduler.py — each
EP rank keeps only the experts the allocator assigns it; the token
shuffle is one uneven all_to_all each way (xGMI all-pairs traffic), with
per-(rank, expert) counts exchanged on a small int tensor first.
"""
from __future__ import annotations

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


# ---------------------------------------------------------------------------
# dynamic expert re-placement
# ---------------------------------------------------------------------------
def rebalance_experts(
    layer: nn.Module,
    token_counts,
    optimizer: Optional[torch.optim.Optimizer] = None,
) -> bool:
    """Dynamic expert re-placement (reference experts_allocator dynamic
    policy, legacy/vescale/moe/experts_allocator.py): all-reduce the
    observed per-expert token counts over the EP group, ask the allocator
    for a new placement, and MOVE re-placed experts' parameters (and, when
    an optimizer is given, their Adam-style state) between ranks with
    point-to-point sends on the EP group.

    The layer must expose `make_expert()` (a fresh expert module on the
    layer's device) — MoELayer does.  Every EP rank must call this with
    its LOCAL counts (list/tensor of length n_experts).  Returns True if
    the placement changed.  If the model also uses a MoEParamBuffer for a
    DP dim, call `buffer.rebuild(model)` afterwards.
    """
    alloc = layer.allocator
    group = layer.ep_group
    if alloc is None:
        return False
    dist_up = dist.is_initialized() and group is not None
    rank = dist.get_rank(group) if dist_up else 0
    counts = torch.as_tensor(token_counts, dtype=torch.float64).clone()
    if dist_up:
        dist.all_reduce(counts, group=group)
    old = [alloc.owner_of(e) for e in range(alloc.n_experts)]
    if not alloc.update(counts.tolist()):
        return False
    new = [alloc.owner_of(e) for e in range(alloc.n_experts)]

    # NOTE: optimizer state transfer covers param_groups[0] and the Adam
    # family keys below — matching how the MoE examples/tests construct
    # the expert optimizer (a single param group)
    opt_keys = ("exp_avg", "exp_avg_sq", "step")
    for e in range(alloc.n_experts):
        if old[e] == new[e]:
            continue
        src_g = dist.get_global_rank(group, old[e]) if dist_up else old[e]
        dst_g = dist.get_global_rank(group, new[e]) if dist_up else new[e]
        if rank == old[e]:
            mod = layer.experts[e]
            for p in mod.parameters():
                dist.send(p.data.contiguous(), dst=dst_g, group=group)
                if optimizer is not None:
                    st = optimizer.state.get(p, {})
                    flags = torch.tensor(
                        [1 if k in st else 0 for k in opt_keys], dtype=torch.int64
                    )
                    dist.send(flags, dst=dst_g, group=group)
                    for k in opt_keys:
                        if k in st:
                            t = st[k]
                            t = t if torch.is_tensor(t) else torch.tensor(float(t))
                            dist.send(
                                t.to(torch.float32).reshape(-1).contiguous(),
                                dst=dst_g, group=group,
                            )
                    optimizer.state.pop(p, None)
            if optimizer is not None:
                plist = optimizer.param_groups[0]["params"]
                mine = set(id(q) for q in mod.parameters())
                optimizer.param_groups[0]["params"] = [
                    q for q in plist if id(q) not in mine
                ]
            layer.experts[e] = _RemoteExpert()
        elif rank == new[e]:
            mod = layer.make_expert()
            for p in mod.parameters():
                buf = torch.empty_like(p.data)
                dist.recv(buf, src=src_g, group=group)
                with torch.no_grad():
                    p.data.copy_(buf)
                if optimizer is not None:
                    flags = torch.empty(len(opt_keys), dtype=torch.int64)
                    dist.recv(flags, src=src_g, group=group)
                    st = {}
                    for j, k in enumerate(opt_keys):
                        if int(flags[j]):
                            n = p.numel() if k != "step" else 1
                            t = torch.empty(n, dtype=torch.float32)
                            dist.recv(t, src=src_g, group=group)
                            st[k] = (
                                t.reshape(p.shape).to(p.device)
                                if k != "step"
                                else t.reshape(()).clone()
                            )
                    optimizer.state[p] = st
            if optimizer is not None:
                optimizer.param_groups[0]["params"].extend(mod.parameters())
            layer.experts[e] = mod
    layer.local_expert_ids = alloc.experts_of(rank)
    return True


def is_experts_parallized(model) -> bool:
    """True if any MoE layer in `model` has been expert-parallelized
    (reference moe/api.py is_experts_parallized; spelling kept)."""
    for mod in model.modules():
        if getattr(mod, "_ep", False):
            return True
    return False
