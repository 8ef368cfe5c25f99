"""Expert placement policy.

Parity: legacy/vescale/moe/experts_allocator.py:26-120 (ExpertsAllocator +
BasicExpertsAllocator): decides which EP rank owns each expert; the Basic
policy is static round-robin; the interface allows DYNAMIC re-placement
(load-driven) between steps — re-placement moves expert params via the
same all-to-all machinery.
"""
from __future__ import annotations

from typing import Dict, List, Optional

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
