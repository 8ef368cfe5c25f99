"""BasicOptimizer — non-ZeRO wrapper coordinating DModule/DDP grad sync.

Parity: legacy/vescale/optim/base_optimizer.py:116.
"""
from __future__ import annotations

from typing import List, Sequence, Union

import torch
import torch.nn as nn

from ..dtensor import DTensor


class BasicOptimizerHook:
    """Grad-sync trigger hook attached to BasicOptimizer (reference
    optim/base_optimizer.py BasicOptimizerHook): subclass and override
    sync() to customize when/how Partial grads are reduced; the default
    calls each registered module's finish_grad_sync()."""

    def sync(self, models) -> None:
        for m in models or []:
            fn = getattr(m, "finish_grad_sync", None)
            if callable(fn):
                fn()


class BasicOptimizer:
    def __init__(
        self,
        optimizer: torch.optim.Optimizer,
        models: Union[nn.Module, Sequence[nn.Module]],
        *,
        grad_hook: bool = True,
    ):
        self.optimizer = optimizer
        self.models: List[nn.Module] = (
            list(models) if isinstance(models, (list, tuple)) else [models]
        )

    def _sync(self):
        for m in self.models:
            target = m.module if hasattr(m, "module") else m
            if hasattr(m, "finish_grad_sync"):
                m.finish_grad_sync()
            elif hasattr(target, "finish_grad_sync"):
                target.finish_grad_sync()

    def step(self, closure=None):
        self._sync()
        # DDP models keep grads in main_grad views — surface them
        for m in self.models:
            for p in m.parameters():
                if getattr(p, "main_grad", None) is not None and p.grad is None:
                    mg = p.main_grad
                    if isinstance(p.data, DTensor):
                        from ..dtensor._dtensor_spec import DTensorSpec
                        from ..dtensor.placement_types import Partial, Replicate

                        spec = p.data._spec
                        pl = tuple(
                            Replicate() if isinstance(q, Partial) else q
                            for q in spec.placements
                        )
                        sp = DTensorSpec(spec.mesh, pl, spec.tensor_meta)
                        p.grad = DTensor(
                            mg.to(p.data._local_tensor.dtype), sp, requires_grad=False
                        )
                    else:
                        p.grad = mg.to(p.data.dtype)
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = True):
        self.optimizer.zero_grad(set_to_none)
        for m in self.models:
            if hasattr(m, "zero_grad_buffer"):
                m.zero_grad_buffer()

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd)

    @property
    def param_groups(self):
        return self.optimizer.param_groups
