"""loss_parallel — cross-entropy on CLASS-dim-sharded logits (vocab
parallel loss).

Parity: legacy/vescale/dtensor/loss.py:39-474 (sharded _log_softmax via
allreduce(max)+allreduce(sum) on the class dim :126-141, fwd/bwd
handlers).  Within the context, F.cross_entropy / F.nll_loss on a
DTensor whose LAST dim is Shard works without gathering the vocab dim —
two scalar-field allreduces replace a full logits all-gather (the xGMI
win: 2 x [N] messages instead of [N, V/w]).
"""
from __future__ import annotations

import contextlib

import torch
import torch.distributed as dist

from . import _collective_utils as cc
from ._dtensor_spec import DTensorSpec
from .dispatch import get_dispatcher
from .placement_types import Partial, Replicate, Shard, TensorMeta

aten = torch.ops.aten


def _class_shard_dim(spec: DTensorSpec, dim: int):
    for md, p in enumerate(spec.placements):
        if isinstance(p, Shard) and p.dim == dim % spec.ndim:
            return md
    return None


def _handler_log_softmax(dispatcher, op, args, kwargs):
    from .dtensor import DTensor
    from .ops.common import out_spec
    from .ops.math_ops import softmax_rule
    from ._op_schema import OpSchema

    x = args[0]
    dim = args[1]
    half_to_float = args[2] if len(args) > 2 else False
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    md = _class_shard_dim(spec, dim)
    if md is None:
        # fall back to the normal rule path
        return dispatcher_default(dispatcher, op, args, kwargs)
    local = x._local_tensor
    xf = local.float() if half_to_float else local
    mx = xf.amax(dim=dim, keepdim=True)
    cc.mesh_all_reduce(mx, spec.mesh, "max", md)
    ex = (xf - mx).exp()
    s = ex.sum(dim=dim, keepdim=True)
    cc.mesh_all_reduce(s, spec.mesh, "sum", md)
    out = xf - mx - s.log()
    tm = TensorMeta(spec.shape, spec.tensor_meta.stride, out.dtype)
    osp = DTensorSpec(spec.mesh, spec.placements, tm)
    return DTensor(out, osp, requires_grad=out.requires_grad)


def _handler_log_softmax_bwd(dispatcher, op, args, kwargs):
    from .dtensor import DTensor

    grad_out, out, dim, in_dtype = args[0], args[1], args[2], args[3]
    if not isinstance(out, DTensor):
        return op(*args, **kwargs)
    spec = out._spec
    md = _class_shard_dim(spec, dim)
    if md is None:
        return dispatcher_default(dispatcher, op, args, kwargs)
    g = grad_out._local_tensor if isinstance(grad_out, DTensor) else grad_out
    o = out._local_tensor
    ssum = g.sum(dim=dim, keepdim=True)
    cc.mesh_all_reduce(ssum, spec.mesh, "sum", md)
    dx = (g - o.exp() * ssum).to(in_dtype)
    tm = TensorMeta(spec.shape, spec.tensor_meta.stride, dx.dtype)
    osp = DTensorSpec(spec.mesh, spec.placements, tm)
    return DTensor(dx, osp, requires_grad=False)


def _handler_nll_forward(dispatcher, op, args, kwargs):
    """nll on class-sharded log-probs: masked local gather + Partial sum."""
    from .dtensor import DTensor
    from .ops.common import out_spec
    from .ops.tensor_ops import _handler_nll_loss_forward

    self_, target = args[0], args[1]
    weight = args[2] if len(args) > 2 else None
    reduction = args[3] if len(args) > 3 else 1
    ignore_index = args[4] if len(args) > 4 else -100
    if not isinstance(self_, DTensor):
        return op(*args, **kwargs)
    spec = self_._spec
    md = _class_shard_dim(spec, -1)
    if md is None:
        return _handler_nll_loss_forward(dispatcher, op, args, kwargs)
    assert weight is None, "loss_parallel does not support class weights"
    mesh = spec.mesh
    local = self_._local_tensor  # [N, V/w] log-probs
    tgt = target._local_tensor if isinstance(target, DTensor) else target
    V = spec.shape[-1]
    w = mesh.size(md)
    r = mesh.get_local_rank(md)
    start = Shard.chunk_offset(V, w, r)
    n = local.shape[-1]
    valid = tgt != ignore_index
    inrange = (tgt >= start) & (tgt < start + n) & valid
    shifted = (tgt - start).clamp(0, max(0, n - 1))
    picked = local.gather(-1, shifted.unsqueeze(-1)).squeeze(-1)
    picked = torch.where(inrange, picked, torch.zeros_like(picked))
    # partial over class shards: each target's logprob lives on one rank
    loss_vec = -picked
    ntok = valid.sum().clamp(min=1).to(local.dtype)
    if reduction == 0:  # none
        pl = [
            Partial("sum") if i == md else (p if not p.is_partial() else Replicate())
            for i, p in enumerate(spec.placements)
        ]
        tm = TensorMeta(torch.Size(spec.shape[:-1]), (1,) * (spec.ndim - 1), loss_vec.dtype)
        osp = DTensorSpec(mesh, tuple(pl), tm)
        tw = torch.zeros((), dtype=local.dtype, device=local.device)
        return DTensor(loss_vec, osp, requires_grad=loss_vec.requires_grad), tw
    total = loss_vec.sum()
    cc.mesh_all_reduce(total, mesh, "sum", md)
    if reduction == 1:
        total = total / ntok
    tm = TensorMeta(torch.Size(()), (), total.dtype)
    pl = tuple(Replicate() for _ in range(mesh.ndim))
    osp = DTensorSpec(mesh, pl, tm)
    return (
        DTensor(total, osp, requires_grad=total.requires_grad),
        DTensor(ntok, osp, requires_grad=False),
    )


def _handler_nll_backward(dispatcher, op, args, kwargs):
    from .dtensor import DTensor
    from .ops.tensor_ops import _handler_nll_loss_backward

    grad_output, self_, target = args[0], args[1], args[2]
    reduction = args[4] if len(args) > 4 else 1
    ignore_index = args[5] if len(args) > 5 else -100
    total_weight = args[6] if len(args) > 6 else None
    if not isinstance(self_, DTensor):
        return op(*args, **kwargs)
    spec = self_._spec
    md = _class_shard_dim(spec, -1)
    if md is None:
        return _handler_nll_loss_backward(dispatcher, op, args, kwargs)
    mesh = spec.mesh
    local = self_._local_tensor
    tgt = target._local_tensor if isinstance(target, DTensor) else target
    g = grad_output._local_tensor if isinstance(grad_output, DTensor) else grad_output
    V = spec.shape[-1]
    w = mesh.size(md)
    r = mesh.get_local_rank(md)
    start = Shard.chunk_offset(V, w, r)
    n = local.shape[-1]
    valid = tgt != ignore_index
    ntok = valid.sum().clamp(min=1).to(local.dtype)
    scale = g / ntok if reduction == 1 else g
    dx = torch.zeros_like(local)
    inrange = (tgt >= start) & (tgt < start + n) & valid
    shifted = (tgt - start).clamp(0, max(0, n - 1))
    src = torch.where(inrange, -scale.expand_as(tgt).to(local.dtype), torch.zeros_like(tgt, dtype=local.dtype))
    dx.scatter_(-1, shifted.unsqueeze(-1), src.unsqueeze(-1))
    tm = spec.tensor_meta
    osp = DTensorSpec(mesh, spec.placements, tm)
    return DTensor(dx, osp, requires_grad=False)


def dispatcher_default(dispatcher, op, args, kwargs):
    """Run the op through the normal rule machinery (bypass this handler)."""
    d = get_dispatcher()
    saved = d._handlers.pop(op, None)
    try:
        return d.dispatch(op, args, kwargs)
    finally:
        if saved is not None:
            d._handlers[op] = saved


@contextlib.contextmanager
def loss_parallel():
    d = get_dispatcher()
    installed = {
        aten._log_softmax.default: _handler_log_softmax,
        aten._log_softmax_backward_data.default: _handler_log_softmax_bwd,
        aten.nll_loss_forward.default: _handler_nll_forward,
        aten.nll_loss_backward.default: _handler_nll_backward,
    }
    saved = {}
    for op, h in installed.items():
        saved[op] = d._handlers.get(op)
        d._handlers[op] = h
    try:
        yield
    finally:
        for op, h in saved.items():
            if h is None:
                d._handlers.pop(op, None)
            else:
                d._handlers[op] = h
