"""Token -> expert-rank dispatch bookkeeping.

Parity: legacy/vescale/moe/token_dispatcher.py:8-90 (TokenDispatcher +
BasicTokenDispatcher): given router top-k assignments, computes the
permutation + per-rank split sizes feeding global_all_to_all_single, and
the inverse combine.
"""
from __future__ import annotations

from typing import List, Tuple

import torch

from .experts_allocator import ExpertsAllocator


class TokenDispatcher:
    def dispatch_plan(self, expert_ids: torch.Tensor):
        raise NotImplementedError


class BasicTokenDispatcher(TokenDispatcher):
    def __init__(self, allocator: ExpertsAllocator):
        self.allocator = allocator

    def dispatch_plan(
        self, expert_ids: torch.Tensor
    ) -> Tuple[torch.Tensor, List[int], torch.Tensor]:
        """expert_ids: [n_tok_k] flat expert assignment (token x top-k
        already flattened).  Returns (perm, send_splits, sorted_experts):
        perm sorts tokens by owning rank (stable, so same-rank tokens stay
        grouped by expert order of appearance); send_splits[r] = tokens
        headed to rank r; sorted_experts = expert ids in permuted order."""
        W = self.allocator.ep_world
        owners = torch.empty_like(expert_ids)
        E = self.allocator.n_experts
        owner_table = torch.tensor(
            [self.allocator.owner_of(e) for e in range(E)],
            device=expert_ids.device,
        )
        owners = owner_table[expert_ids]
        perm = torch.argsort(owners * (E + 1) + expert_ids, stable=True)
        send_splits = torch.bincount(owners, minlength=W).tolist()
        sorted_experts = expert_ids[perm]
        return perm, send_splits, sorted_experts
