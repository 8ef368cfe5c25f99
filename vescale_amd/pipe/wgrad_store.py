"""Deferred weight-gradient store for zero-bubble pipeline schedules.

Parity: the reference's zero-bubble W/B split
(legacy/vescale/pipe/_schedules/zero_bubble_v.py) separates each
microbatch's backward into B (input grads — on the pipeline critical
path) and W (weight grads — bubble filler).  This store implements the
TRUE split: during a B-phase backward, every patched nn.Linear computes
only dX and pushes a closure computing dW (and db) from its saved
(x, gy); the W-phase instruction pops and runs the closures.  No second
graph traversal and no retain_graph — the old double-traversal form
(correct but ~2x backward GEMM work) is replaced.
"""
from __future__ import annotations

import types
from typing import Callable, Dict, List, Tuple

import torch
import torch.nn as nn


class WeightGradStore:
    """Collects deferred weight-grad closures per (chunk, microbatch)."""

    def __init__(self) -> None:
        self._active = False
        self._cur: List[Callable[[], None]] = []
        self._by_key: Dict[Tuple[int, int], List[Callable[[], None]]] = {}

    def begin(self) -> None:
        self._active = True
        self._cur = []

    def end(self, key: Tuple[int, int]) -> None:
        self._by_key[key] = self._cur
        self._active = False
        self._cur = []

    def defer(self, fn: Callable[[], None]) -> bool:
        """Queue fn if a B-phase is active; returns False to run inline."""
        if self._active:
            self._cur.append(fn)
            return True
        return False

    def pop_run(self, key: Tuple[int, int]) -> None:
        # W-phase grad GEMMs must not record autograd history (they run in
        # normal grad mode from the instruction loop, unlike Function.backward)
        with torch.no_grad():
            for fn in self._by_key.pop(key, []):
                fn()


def _accum_grad(param: torch.Tensor, g: torch.Tensor) -> None:
    """Accumulate a deferred W-phase gradient the way autograd would have.

    When the stage is DDP-wrapped, grads must land in the flat main_grad
    buffer AND register bucket readiness — writing param.grad directly would
    silently drop the grad from DP reduction.  DDP exposes its param hook as
    `param._ddp_param_hook` (ddp/distributed_data_parallel.py) for exactly
    this producer-outside-autograd case.
    """
    hook = getattr(param, "_ddp_param_hook", None)
    if param.grad is None:
        param.grad = g
    else:
        param.grad = param.grad + g
    if hook is not None:
        hook(param)  # moves .grad into main_grad, clears it, registers ready


class _ZBLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, store, module):
        ctx.save_for_backward(x)
        ctx.store = store
        ctx.module = module            # nn.Linear: grad sink (non-tensor arg)
        ctx.w_value = weight.detach()  # value for dX
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, gy):
        (x,) = ctx.saved_tensors
        w = ctx.w_value
        gx = gy.matmul(w)  # [.., out] x [out, in] -> [.., in]
        w_param = ctx.module.weight
        b_param = ctx.module.bias

        def compute_wgrad(gy=gy, x=x, w_param=w_param, b_param=b_param):
            g2 = gy.reshape(-1, gy.shape[-1])
            x2 = x.reshape(-1, x.shape[-1])
            gw = g2.t().matmul(x2)
            _accum_grad(w_param, gw)
            if b_param is not None:
                _accum_grad(b_param, g2.sum(0))

        if not ctx.store.defer(compute_wgrad):
            compute_wgrad()
        return gx, None, None, None, None


def zb_patch_linears(module: nn.Module, store: WeightGradStore) -> int:
    """Route every nn.Linear in `module` through the deferring autograd
    Function.  Idempotent; returns the number of layers patched."""
    n = 0
    for m in module.modules():
        if isinstance(m, nn.Linear) and not getattr(m, "_zb_patched", False):

            def fwd(self, x, _store=store):
                return _ZBLinearFn.apply(x, self.weight, self.bias, _store, self)

            m.forward = types.MethodType(fwd, m)
            m._zb_patched = True
            n += 1
    return n
