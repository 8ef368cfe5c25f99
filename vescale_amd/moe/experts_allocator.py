"""Expert placement policy.

Parity: legacy/vescale/moe/experts_allocator.py:26-120 (ExpertsAllocator +
BasicExpertsAllocator): decides which EP rank owns each expert; the Basic
policy is static round-robin; the interface allows DYNAMIC re-placement
(load-driven) between steps — re-placement moves expert params via the
same all-to-all machinery.
"""
from __future__ import annotations

from typing import List, Optional

import torch.distributed as dist


class ExpertsAllocator:
    """Interface: expert_id -> owning EP rank (may change over time)."""

    def __init__(self, n_experts: int, ep_world: int):
        self.n_experts = n_experts
        self.ep_world = ep_world

    def owner_of(self, expert_id: int) -> int:
        raise NotImplementedError

    def experts_of(self, rank: int) -> List[int]:
        return [e for e in range(self.n_experts) if self.owner_of(e) == rank]

    def update(self, token_counts: Optional[List[int]] = None):
        """Hook for dynamic re-placement from observed per-expert load."""
        return False  # no change


class BasicExpertsAllocator(ExpertsAllocator):
    """Static blocked placement: expert e lives on rank e // (E/W)."""

    def owner_of(self, expert_id: int) -> int:
        per = max(1, self.n_experts // self.ep_world)
        return min(expert_id // per, self.ep_world - 1)


class LoadBalancedExpertsAllocator(ExpertsAllocator):
    """DYNAMIC placement (reference's dynamic allocator policy): starts
    blocked like Basic, and `update(token_counts)` re-places experts with
    a greedy longest-processing-time bin packing over the observed load,
    capped at ceil(E/W) experts per rank (bounds per-rank memory).
    Returns True when the placement changed; the caller moves the params
    (moe.api.rebalance_experts)."""

    def __init__(self, n_experts: int, ep_world: int):
        super().__init__(n_experts, ep_world)
        per = max(1, n_experts // ep_world)
        self.placement: List[int] = [
            min(e // per, ep_world - 1) for e in range(n_experts)
        ]

    def owner_of(self, expert_id: int) -> int:
        return self.placement[expert_id]

    def update(self, token_counts: Optional[List[int]] = None) -> bool:
        if not token_counts or self.ep_world <= 1:
            return False
        cap = -(-self.n_experts // self.ep_world)  # ceil(E/W)
        order = sorted(
            range(self.n_experts), key=lambda e: -float(token_counts[e])
        )
        load = [0.0] * self.ep_world
        slots = [0] * self.ep_world
        new: List[int] = [0] * self.n_experts
        for e in order:
            # least-loaded rank with a free slot; ties -> keep current owner
            cands = [r for r in range(self.ep_world) if slots[r] < cap]
            cands.sort(key=lambda r: (load[r], r != self.placement[e]))
            r = cands[0]
            new[e] = r
            load[r] += float(token_counts[e])
            slots[r] += 1
        if new == self.placement:
            return False
        self.placement = new
        return True
