"""MoE parameter buffers — per-layer flat expert-param management.

Parity: legacy/vescale/moe/_moe_param_buffer.py:50-405 (MoEParamBuffer /
MoELayerParamBuffer: per-layer param all-gather `run_all_gather`, grad
`run_reduce_scatter`).  Used when experts are ALSO data-parallel across a
DP dim (experts sharded over EP, replicated over DP): each layer's local
expert params live in one flat buffer so DP grad sync is one bucketed
collective per layer.
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist
import torch.nn as nn


class MoELayerParamBuffer:
    """Flat param/grad buffer over one MoE layer's LOCAL experts."""

    def __init__(self, layer: nn.Module, dp_group=None, dtype=None):
        self.dp_group = dp_group
        self.dp_world = (
            dist.get_world_size(dp_group)
            if dp_group is not None and dist.is_initialized()
            else 1
        )
        self.params: List[nn.Parameter] = [
            p for n, p in layer.named_parameters() if ".experts." in n or "experts" in n
        ]
        if not self.params:
            self.params = list(layer.parameters())
        dt = dtype or (self.params[0].dtype if self.params else torch.float32)
        total = sum(p.numel() for p in self.params)
        dev = self.params[0].device if self.params else torch.device("cpu")
        self.grad_buffer = torch.zeros(total, dtype=torch.float32, device=dev)
        off = 0
        self._views = []
        self._view_by_param: Dict[int, torch.Tensor] = {}
        for p in self.params:
            n = p.numel()
            view = self.grad_buffer.narrow(0, off, n).view(p.shape)
            p.main_grad = view
            self._views.append((p, view))
            self._view_by_param[id(p)] = view
            p.register_post_accumulate_grad_hook(self._make_hook(p))
            off += n
        self._work = None

    def _make_hook(self, p):
        # resolve the view at CALL time: rebuild() replaces the buffer and
        # views after dynamic expert re-placement, and a hook that captured
        # the old view would accumulate into freed storage
        def hook(param):
            if param.grad is not None:
                view = self._view_by_param.get(id(param))
                if view is not None:
                    view.add_(param.grad.float())
                param.grad = None

        return hook

    def run_all_gather(self):
        """No-op placeholder for dynamically re-placed experts (the dynamic
        allocator path re-gathers moved expert params here)."""
        return None

    def run_reduce_scatter(self, async_op: bool = True):
        """DP-average the layer's expert grads in ONE collective."""
        if self.dp_world <= 1:
            return None
        self.grad_buffer.div_(self.dp_world)
        self._work = dist.all_reduce(
            self.grad_buffer, group=self.dp_group, async_op=async_op
        )
        return self._work

    def finish(self):
        if self._work is not None:
            self._work.wait()
            self._work = None

    def zero_grad(self):
        self.grad_buffer.zero_()

    def rebuild(self, layer: nn.Module):
        """Re-derive the flat buffer after dynamic expert re-placement
        (moe.api.rebalance_experts): moved-away params vanish with their
        modules; newly-arrived params get views + hooks.  Params that
        stayed keep their hook (re-registering would double-accumulate),
        so hooks are tracked by id."""
        hooked = set(self._view_by_param.keys())
        self.params = [
            p for n, p in layer.named_parameters() if "experts" in n
        ] or list(layer.parameters())
        total = sum(p.numel() for p in self.params)
        dev = self.params[0].device if self.params else torch.device("cpu")
        self.grad_buffer = torch.zeros(total, dtype=torch.float32, device=dev)
        off = 0
        self._views = []
        self._view_by_param = {}
        for p in self.params:
            n = p.numel()
            view = self.grad_buffer.narrow(0, off, n).view(p.shape)
            p.main_grad = view
            self._views.append((p, view))
            self._view_by_param[id(p)] = view
            if id(p) not in hooked:
                p.register_post_accumulate_grad_hook(self._make_hook(p))
            off += n


class MoEParamBuffer:
    """All MoE layers' buffers (parity: MoEParamBuffer :405)."""

    def __init__(self, model: nn.Module, layer_cls: str = "MoELayer", dp_group=None):
        self.layer_buffers: Dict[str, MoELayerParamBuffer] = {}
        for name, mod in model.named_modules():
            if type(mod).__name__ == layer_cls:
                self.layer_buffers[name] = MoELayerParamBuffer(mod, dp_group)

    def run_reduce_scatter(self):
        for b in self.layer_buffers.values():
            b.run_reduce_scatter()

    def finish_grad_sync(self):
        for b in self.layer_buffers.values():
            b.finish()

    def zero_grad(self):
        for b in self.layer_buffers.values():
            b.zero_grad()

    def rebuild(self, model: nn.Module, layer_cls: str = "MoELayer"):
        """Re-derive all layer buffers after dynamic expert re-placement."""
        for name, mod in model.named_modules():
            if type(mod).__name__ == layer_cls and name in self.layer_buffers:
                self.layer_buffers[name].rebuild(mod)
