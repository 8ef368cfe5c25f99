"""PipeEngine — the user-facing pipeline training engine.

Parity: legacy/vescale/engine/pipe.py:51-255 (build_schedule,
forward_backward, parameters, sync_shared_params).
"""
from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch
import torch.distributed as dist

from ..plan import PipelineParallelPlan
from ..pipe.pipe_emmiter import ScheduleEngine
from ..pipe.pipe_stage import PipeModule


class PipeEngine:
    def __init__(
        self,
        stage: PipeModule,
        plan: PipelineParallelPlan,
        *,
        loss_fn: Optional[Callable] = None,
        stage_to_rank: Optional[Callable[[int], int]] = None,
        pg=None,
        device: Optional[torch.device] = None,
    ):
        self.module = stage
        self.plan = plan
        self.schedule_engine = ScheduleEngine(
            stage, plan, stage_to_rank=stage_to_rank, pg=pg, loss_fn=loss_fn,
            device=device,
        )

    def build_schedule(self, n_microbatches: int):
        return self.schedule_engine.build_schedule(n_microbatches)

    def forward_backward(
        self,
        minibatch: Optional[Tuple[torch.Tensor, torch.Tensor]],
        n_microbatches: int,
        *,
        forward_only: bool = False,
    ):
        """Returns the summed minibatch loss on the LAST stage (None
        elsewhere); gradients are left on stage parameters.

        forward_only=True (or calling under torch.no_grad()) runs the
        eval projection of the schedule: forwards and activation sends
        only, no grads (reference _schedules forward_only)."""
        return self.schedule_engine.execute(
            minibatch, n_microbatches, forward_only=forward_only
        )

    def evaluate(
        self,
        minibatch: Optional[Tuple[torch.Tensor, torch.Tensor]],
        n_microbatches: int,
    ):
        """Validation-loop convenience: forward-only, no autograd state."""
        return self.forward_backward(minibatch, n_microbatches, forward_only=True)

    def parameters(self):
        return self.module.parameters()

    def sync_shared_params(self):
        self.module.sync_shared_params()

    def zero_grad(self):
        for p in self.module.parameters():
            p.grad = None
