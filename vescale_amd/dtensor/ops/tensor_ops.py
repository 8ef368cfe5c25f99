"""Tensor ops: factories-like, index ops, bypasses, foreach/fused handlers.

Parity: legacy/vescale/dtensor/ops/tensor_ops.py, _dispatch_bypass.py, and
the fused-optimizer unwrap at vescale/dtensor/_dispatch.py:118,269-271.
"""
from __future__ import annotations


import torch
import torch.distributed as dist

from .._dtensor_spec import DTensorSpec
from .._op_schema import OpSchema, OutputSharding
from ..placement_types import (
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
)
from .common import out_spec, pointwise_rule, same_as_input_rule

aten = torch.ops.aten


# ---------------------------------------------------------------------------
# *_like factories
# ---------------------------------------------------------------------------
def zeros_like_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    # zero is the identity of sum: Partial stays Partial (harmless), shards stay
    dtype = s.dtype
    kd = schema.kwargs_schema.get("dtype")
    if isinstance(kd, torch.dtype):
        dtype = kd
    return OutputSharding(out_spec(s.mesh, s.placements, tuple(s.shape), dtype), None)


def nonzero_preserving_like_rule(schema: OpSchema) -> OutputSharding:
    """ones_like/full_like: Partial input would produce wrong global value;
    emit Replicate there instead (value is position-independent)."""
    s = schema.specs[0]
    pl = [Replicate() if isinstance(p, Partial) else p for p in s.placements]
    dtype = s.dtype
    kd = schema.kwargs_schema.get("dtype")
    if isinstance(kd, torch.dtype):
        dtype = kd
    return OutputSharding(out_spec(s.mesh, pl, tuple(s.shape), dtype), None)


# ---------------------------------------------------------------------------
# index ops
# ---------------------------------------------------------------------------
def index_select_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] % s.ndim
    idx = schema.specs[1] if len(schema.specs) > 1 else None
    targets_x, pl = [], []
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, (RaggedShard, Partial)):
            targets_x.append(Replicate())
            pl.append(Replicate())
        else:
            targets_x.append(p)
            pl.append(p)
    rep = tuple(Replicate() for _ in range(s.mesh.ndim))
    shape = list(s.shape)
    if idx is not None:
        shape[d] = idx.shape[0]
    osp = out_spec(s.mesh, pl, shape, s.dtype)
    targets = [tuple(targets_x)] + ([rep] if idx is not None else [])
    return OutputSharding(osp, targets)


def gather_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] % s.ndim
    idx = schema.specs[2] if len(schema.specs) > 2 else schema.specs[1]
    rep = tuple(Replicate() for _ in range(s.mesh.ndim))
    # conservative: replicate both unless sharded on a non-gather dim equally
    osp = out_spec(s.mesh, rep, tuple(idx.shape), s.dtype)
    return OutputSharding(osp, [rep for _ in schema.specs])


def scatter_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    rep = tuple(Replicate() for _ in range(s.mesh.ndim))
    osp = out_spec(s.mesh, rep, tuple(s.shape), s.dtype)
    return OutputSharding(osp, [rep for _ in schema.specs])


# ---------------------------------------------------------------------------
# bypasses: answered without touching placement logic
# ---------------------------------------------------------------------------
def _bypass_equal(op, args, kwargs):
    from ..dtensor import DTensor

    a, b = args[0], args[1]
    if isinstance(a, DTensor) and isinstance(b, DTensor):
        if a._spec.placements != b._spec.placements:
            b = b.redistribute(placements=a._spec.placements)
        local_eq = torch.equal(a._local_tensor, b._local_tensor)
        t = torch.tensor([0 if local_eq else 1], dtype=torch.int64)
        if a._spec.mesh.size() > 1 and dist.is_initialized():
            for md in range(a._spec.mesh.ndim):
                dist.all_reduce(t, group=a._spec.mesh.get_group(md))
        return bool(t.item() == 0)
    return NotImplemented


def _bypass_is_same_size(op, args, kwargs):
    a, b = args
    return tuple(a.shape) == tuple(b.shape)


def _bypass_local_scalar_dense(op, args, kwargs):
    from ..dtensor import DTensor

    (a,) = args
    assert isinstance(a, DTensor)
    spec = a._spec
    local = a._local_tensor
    for md, p in enumerate(spec.placements):
        if isinstance(p, Partial):
            local = local.clone()
            from .. import _collective_utils as cc

            cc.mesh_all_reduce(local, spec.mesh, p.reduce_op, md)
        elif isinstance(p, (Shard, InterleavedShard, RaggedShard)):
            raise RuntimeError(".item() on a sharded DTensor is ambiguous; redistribute first")
    return op(local)


# ---------------------------------------------------------------------------
# handlers needing eager logic
# ---------------------------------------------------------------------------
def _handler_foreach_pointwise(dispatcher, op, args, kwargs):
    """_foreach_* ops: unwrap every DTensor in the lists; placements are
    preserved per-element.  Mixing shardings within a list is allowed (each
    element is independent)."""
    from ..dtensor import DTensor

    def unwrap(x):
        if isinstance(x, DTensor):
            return x._local_tensor
        if isinstance(x, (list, tuple)):
            return type(x)(unwrap(y) for y in x)
        return x

    local_args = tuple(unwrap(a) for a in args)
    local_kwargs = {k: unwrap(v) for k, v in kwargs.items()}
    res = op(*local_args, **local_kwargs)
    if res is None:
        return None
    # wrap outputs following the first list's specs
    first_list = None
    for a in args:
        if isinstance(a, (list, tuple)) and any(isinstance(x, DTensor) for x in a):
            first_list = a
            break
    if first_list is None or not isinstance(res, (list, tuple)):
        return res
    out = []
    for r, ref in zip(res, first_list):
        if isinstance(ref, DTensor) and isinstance(r, torch.Tensor):
            tm = TensorMeta(ref._spec.shape, ref._spec.tensor_meta.stride, r.dtype)
            sp = DTensorSpec(ref._spec.mesh, ref._spec.placements, tm)
            out.append(DTensor(r, sp, requires_grad=r.requires_grad))
        else:
            out.append(r)
    return type(res)(out)


def _handler_foreach_norm(dispatcher, op, args, kwargs):
    """_foreach_norm on sharded/ragged tensors: local norms are NOT the
    global norms.  Return per-tensor Partial-friendly result: we compute
    local norm and mark Partial via the "norm" trick: callers (grad clip)
    must combine as sqrt(allreduce(sum(local^2))).  To stay safe in generic
    use, we reduce eagerly here (scalar allreduce per call, batched)."""
    from ..dtensor import DTensor

    tensors = args[0]
    ordp = args[1] if len(args) > 1 else 2
    locals_ = [t._local_tensor if isinstance(t, DTensor) else t for t in tensors]
    local_norms = torch._foreach_norm(locals_, ordp)
    # identify sharded elements
    mesh = None
    for t in tensors:
        if isinstance(t, DTensor):
            mesh = t._spec.mesh
            break
    if mesh is None or mesh.size() == 1:
        outs = []
        for t, n in zip(tensors, local_norms):
            if isinstance(t, DTensor):
                sp = out_spec(t._spec.mesh, [Replicate()] * t._spec.mesh.ndim, (), n.dtype)
                outs.append(DTensor(n, sp, requires_grad=False))
            else:
                outs.append(n)
        return outs
    stacked = torch.stack([n.pow(2) for n in local_norms])
    for md in range(mesh.ndim):
        needs = any(
            isinstance(t, DTensor)
            and not t._spec.placements[md].is_replicate()
            for t in tensors
        )
        if needs:
            dist.all_reduce(stacked, group=mesh.get_group(md))
    global_norms = stacked.sqrt()
    outs = []
    for i, t in enumerate(tensors):
        n = global_norms[i]
        if isinstance(t, DTensor):
            sp = out_spec(t._spec.mesh, [Replicate()] * t._spec.mesh.ndim, (), n.dtype)
            outs.append(DTensor(n, sp, requires_grad=False))
        else:
            outs.append(n)
    return outs


def _handler_amp_found_inf(dispatcher, op, args, kwargs):
    """aten._amp_foreach_non_finite_check_and_unscale_: the GradScaler inf
    scan.  Unscale each local shard in place, then combine found_inf with a
    MAX all-reduce across every mesh dim on which any grad is non-replicate —
    an inf seen by ONE rank must poison the step on ALL ranks (reference
    found_inf_reduce_handler, vescale/dtensor/_dispatch.py:60-117, which
    wraps found_inf as Partial("max") and redistributes to Replicate)."""
    from ..dtensor import DTensor

    grads, found_inf, inv_scale = args[0], args[1], args[2]
    locals_ = [g._local_tensor if isinstance(g, DTensor) else g for g in grads]
    fi = found_inf._local_tensor if isinstance(found_inf, DTensor) else found_inf
    sc = inv_scale._local_tensor if isinstance(inv_scale, DTensor) else inv_scale
    op(locals_, fi, sc)
    mesh = next((g._spec.mesh for g in grads if isinstance(g, DTensor)), None)
    if mesh is not None and dist.is_initialized():
        for md in range(mesh.ndim):
            if mesh.size(md) <= 1:
                continue
            needs = any(
                isinstance(g, DTensor)
                and not g._spec.placements[md].is_replicate()
                for g in grads
            )
            if needs:
                dist.all_reduce(fi, op=dist.ReduceOp.MAX, group=mesh.get_group(md))
    return None


def _handler_fused_adam(dispatcher, op, args, kwargs):
    """aten._fused_adamw_/_fused_adam_/_fused_sgd_: unwrap DTensor lists so
    the fused multi-tensor kernel runs directly on local shards (reference
    vescale/dtensor/_dispatch.py:118)."""
    from ..dtensor import DTensor

    def unwrap(x):
        if isinstance(x, DTensor):
            return x._local_tensor
        if isinstance(x, (list, tuple)):
            return type(x)(unwrap(y) for y in x)
        return x

    local_args = tuple(unwrap(a) for a in args)
    local_kwargs = {k: unwrap(v) for k, v in kwargs.items()}
    op(*local_args, **local_kwargs)
    return None


def _handler_vector_norm(dispatcher, op, args, kwargs):
    """linalg_vector_norm over sharded tensors: local p-norm^p -> allreduce
    -> root.  Returns a Replicate scalar DTensor."""
    from ..dtensor import DTensor

    a = args[0]
    if not isinstance(a, DTensor):
        return op(*args, **kwargs)
    ordp = args[1] if len(args) > 1 else 2.0
    spec = a._spec
    sharded_dims = [
        md for md, p in enumerate(spec.placements) if not p.is_replicate()
    ]
    partial_dims = [md for md, p in enumerate(spec.placements) if p.is_partial()]
    if partial_dims:
        a = a.redistribute(placements=[
            Replicate() if p.is_partial() else p for p in spec.placements
        ])
        spec = a._spec
        sharded_dims = [md for md, p in enumerate(spec.placements) if not p.is_replicate()]
    local = a._local_tensor
    if not sharded_dims or spec.mesh.size() == 1:
        res = op(local, *args[1:], **kwargs)
    else:
        if ordp == float("inf"):
            res = local.abs().max() if local.numel() else local.new_zeros(())
            for md in sharded_dims:
                dist.all_reduce(res, op=dist.ReduceOp.MAX, group=spec.mesh.get_group(md))
        else:
            res = local.abs().pow(ordp).sum()
            for md in sharded_dims:
                dist.all_reduce(res, group=spec.mesh.get_group(md))
            res = res.pow(1.0 / ordp)
    sp = out_spec(spec.mesh, [Replicate()] * spec.mesh.ndim, (), res.dtype)
    return DTensor(res, sp, requires_grad=res.requires_grad)


def _handler_nll_loss_forward(dispatcher, op, args, kwargs):
    """nll_loss_forward with batch-sharded input: local sums + Partial;
    mean computed as global sum / global weight-count."""
    from ..dtensor import DTensor

    self_, target = args[0], args[1]
    weight = args[2] if len(args) > 2 else None
    reduction = args[3] if len(args) > 3 else 1
    ignore_index = args[4] if len(args) > 4 else -100
    if not isinstance(self_, DTensor):
        return op(*args, **kwargs)
    spec = self_._spec
    mesh = spec.mesh
    # class dim must be replicate here (loss_parallel handles class sharding)
    batch_sharded_dims = [
        md for md, p in enumerate(spec.placements)
        if isinstance(p, (Shard, InterleavedShard)) and p.dim == 0
    ]
    bad = any(
        (isinstance(p, (Shard, InterleavedShard)) and p.dim != 0) or isinstance(p, (RaggedShard, Partial))
        for p in spec.placements
    )
    if bad:
        rep = [Replicate()] * mesh.ndim
        self_ = self_.redistribute(placements=rep)
        if isinstance(target, DTensor):
            target = target.redistribute(placements=rep)
        spec = self_._spec
        batch_sharded_dims = []
    local_x = self_._local_tensor
    local_t = target._local_tensor if isinstance(target, DTensor) else target
    local_w = weight._local_tensor if isinstance(weight, DTensor) else weight
    # run local with reduction=sum, then combine
    red_local = 2 if reduction != 0 else 0  # sum or none
    out, total_weight = op(local_x, local_t, local_w, red_local, ignore_index)
    if reduction == 0:
        pl = spec.placements
        osp = out_spec(mesh, pl, (spec.shape[0],), out.dtype)
        return (
            DTensor(out, osp, requires_grad=out.requires_grad),
            DTensor(total_weight, out_spec(mesh, [Replicate()] * mesh.ndim, (), total_weight.dtype), requires_grad=False),
        )
    if batch_sharded_dims:
        pl = [Partial("sum") if md in batch_sharded_dims else Replicate() for md in range(mesh.ndim)]
    else:
        pl = [Replicate()] * mesh.ndim
    out_d = DTensor(out, out_spec(mesh, pl, (), out.dtype), requires_grad=out.requires_grad)
    tw_d = DTensor(
        total_weight.clone(), out_spec(mesh, pl, (), total_weight.dtype), requires_grad=False
    )
    if reduction == 1:  # mean = sum / total_weight (both partial-summed)
        out_r = out_d.redistribute(placements=[Replicate()] * mesh.ndim)
        tw_r = tw_d.redistribute(placements=[Replicate()] * mesh.ndim)
        mean = out_r / torch.clamp(tw_r, min=1e-12) if False else out_r / tw_r
        return mean, tw_r
    return out_d, tw_d


def _handler_nll_loss_backward(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    (grad_output, self_, target, weight, reduction, ignore_index, total_weight) = args[:7]
    if not isinstance(self_, DTensor):
        return op(*args, **kwargs)
    spec = self_._spec
    mesh = spec.mesh

    def loc(x):
        return x._local_tensor if isinstance(x, DTensor) else x

    g = loc(grad_output)
    tw = loc(total_weight)
    if reduction == 1:
        # forward returned global mean: d/dx mean = g / total_weight with
        # reduction=sum semantics locally
        local = op(
            g / tw if tw.numel() else g, loc(self_), loc(target), loc(weight), 2, ignore_index,
            torch.ones_like(tw),
        )
    else:
        local = op(g, loc(self_), loc(target), loc(weight), reduction, ignore_index, tw)
    osp = out_spec(mesh, spec.placements, tuple(spec.shape), local.dtype)
    return DTensor(local, osp, requires_grad=local.requires_grad)


def where_rule(schema: OpSchema) -> OutputSharding:
    return pointwise_rule(schema)


def arange_like_factory_bypass(op, args, kwargs):
    return NotImplemented


def register(dispatcher):
    dispatcher.register_rule(aten.zeros_like.default, zeros_like_rule)
    dispatcher.register_rule(aten.empty_like.default, zeros_like_rule)
    dispatcher.register_rule(aten.ones_like.default, nonzero_preserving_like_rule)
    dispatcher.register_rule(aten.full_like.default, nonzero_preserving_like_rule)
    dispatcher.register_rule(aten.index_select.default, index_select_rule)
    dispatcher.register_rule(aten.gather.default, gather_rule)
    dispatcher.register_rule(aten.scatter.src, scatter_rule)
    dispatcher.register_rule(aten.scatter.value, scatter_rule)
    dispatcher.register_rule(aten.scatter_add.default, scatter_rule)

    dispatcher.register_bypass(aten.equal.default, _bypass_equal)
    dispatcher.register_bypass(aten.is_same_size.default, _bypass_is_same_size)
    dispatcher.register_bypass(aten._local_scalar_dense.default, _bypass_local_scalar_dense)

    # foreach family
    for name in dir(aten):
        if name.startswith("_foreach_") and name != "_foreach_norm":
            try:
                dispatcher.register_handler(getattr(aten, name), _handler_foreach_pointwise)
            except (AttributeError, RuntimeError):
                pass
    dispatcher.register_handler(aten._foreach_norm, _handler_foreach_norm)
    for op in (aten._fused_adamw_, aten._fused_adam_, aten._fused_sgd_):
        dispatcher.register_handler(op, _handler_fused_adam)
    if hasattr(aten, "_amp_foreach_non_finite_check_and_unscale_"):
        dispatcher.register_handler(
            aten._amp_foreach_non_finite_check_and_unscale_,
            _handler_amp_found_inf,
        )
    dispatcher.register_handler(aten.linalg_vector_norm.default, _handler_vector_norm)
    dispatcher.register_handler(aten.nll_loss_forward.default, _handler_nll_loss_forward)
    dispatcher.register_handler(aten.nll_loss_backward.default, _handler_nll_loss_backward)
