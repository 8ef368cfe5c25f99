"""Reduction / normalization / softmax rules (parity: legacy/vescale/
dtensor/ops/math_ops.py, vescale/dtensor/_ops/_math_ops.py)."""
from __future__ import annotations

from typing import List, Optional

import torch

from .._dtensor_spec import DTensorSpec
from .._op_schema import OpSchema, OutputSharding
from ..placement_types import (
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
)
from .common import out_spec

aten = torch.ops.aten


def _reduced_shape(shape, dims, keepdim):
    if dims is None:
        return () if not keepdim else tuple(1 for _ in shape)
    out = []
    for i, s in enumerate(shape):
        if i in dims:
            if keepdim:
                out.append(1)
        else:
            out.append(s)
    return tuple(out)


def _reduction_rule(schema: OpSchema, reduce_kind: str) -> OutputSharding:
    s = schema.specs[0]
    mesh = s.mesh
    args = schema.args_schema
    # parse dims/keepdim by overload
    dims: Optional[List[int]] = None
    keepdim = False
    if len(args) > 1 and isinstance(args[1], (list, tuple)):
        dims = [d % s.ndim for d in args[1]]
        if len(args) > 2 and isinstance(args[2], bool):
            keepdim = args[2]
    elif len(args) > 1 and isinstance(args[1], int):
        dims = [args[1] % s.ndim]
        if len(args) > 2 and isinstance(args[2], bool):
            keepdim = args[2]
    all_dims = dims is None
    rdims = set(range(s.ndim)) if all_dims else set(dims)

    # dim remap for output
    if keepdim or all_dims:
        dim_map = {i: i for i in range(s.ndim)}
    else:
        dim_map = {}
        new = 0
        for i in range(s.ndim):
            if i not in rdims:
                dim_map[i] = new
                new += 1

    targets = [list(s.placements)]
    out_pl: List[Placement] = []
    uneven_mean = False
    for md, p in enumerate(s.placements):
        if isinstance(p, RaggedShard):
            # reducing a ragged tensor over everything -> Partial
            if all_dims and reduce_kind != "mean":
                out_pl.append(Partial(reduce_kind))
            elif all_dims and len(set(p.local_units)) == 1:
                # equal units: mean of equal-sized local means == global mean
                # (Partial("sum") here summed the local means — caught by
                # the ragged op sweep)
                out_pl.append(Partial("avg"))
            else:
                # uneven mean (or dim-reduction): no Partial combine exists
                targets[0][md] = Replicate()
                out_pl.append(Replicate())
            continue
        if isinstance(p, (Shard, InterleavedShard)):
            if p.dim in rdims:
                if reduce_kind == "sum":
                    out_pl.append(Partial("sum"))
                elif reduce_kind == "mean":
                    total = s.shape[p.dim]
                    w = mesh.size(md)
                    even = total % w == 0 and not isinstance(p, InterleavedShard)
                    if even:
                        out_pl.append(Partial("avg"))
                    else:
                        targets[0][md] = Replicate()
                        out_pl.append(Replicate())
                elif reduce_kind in ("max", "min"):
                    out_pl.append(Partial(reduce_kind))
                else:
                    targets[0][md] = Replicate()
                    out_pl.append(Replicate())
            else:
                nd = dim_map.get(p.dim)
                if nd is None:
                    targets[0][md] = Replicate()
                    out_pl.append(Replicate())
                elif isinstance(p, InterleavedShard):
                    out_pl.append(InterleavedShard(nd, p.interleaved_size))
                else:
                    out_pl.append(Shard(nd))
        elif isinstance(p, Partial):
            if reduce_kind in ("sum", "mean"):
                out_pl.append(p)  # sum of partial sums commutes
            else:
                targets[0][md] = Replicate()
                out_pl.append(Replicate())
        else:
            out_pl.append(Replicate())

    shape = _reduced_shape(tuple(s.shape), None if all_dims else rdims, keepdim)
    dtype = s.dtype
    kd = schema.kwargs_schema.get("dtype")
    if isinstance(kd, torch.dtype):
        dtype = kd
    osp = out_spec(mesh, out_pl, shape, dtype)
    return OutputSharding(osp, [tuple(targets[0])])


def sum_rule(schema):
    return _reduction_rule(schema, "sum")


def mean_rule(schema):
    return _reduction_rule(schema, "mean")


def amax_rule(schema):
    return _reduction_rule(schema, "max")


def amin_rule(schema):
    return _reduction_rule(schema, "min")


def softmax_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] % s.ndim
    targets = []
    pl = []
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, (RaggedShard, Partial)):
            targets.append(Replicate())
            pl.append(Replicate())
        else:
            targets.append(p)
            pl.append(p)
    osp = out_spec(s.mesh, pl, tuple(s.shape), s.dtype)
    return OutputSharding(osp, [tuple(targets)])


def softmax_bwd_rule(schema: OpSchema) -> OutputSharding:
    g, out = schema.specs[0], schema.specs[1]
    d = schema.args_schema[2] % g.ndim
    targets_g, targets_o, pl = [], [], []
    for pg, po in zip(g.placements, out.placements):
        bad_g = (isinstance(pg, (Shard, InterleavedShard)) and pg.dim == d) or isinstance(pg, (RaggedShard, Partial))
        if bad_g:
            targets_g.append(Replicate())
            targets_o.append(Replicate())
            pl.append(Replicate())
        else:
            targets_g.append(pg)
            targets_o.append(pg)
            pl.append(pg)
    osp = out_spec(g.mesh, pl, tuple(g.shape), g.dtype)
    return OutputSharding(osp, [tuple(targets_g), tuple(targets_o)])


def layer_norm_rule(schema: OpSchema) -> OutputSharding:
    # native_layer_norm(input, normalized_shape, weight, bias, eps)
    s = schema.specs[0]
    normalized_shape = schema.args_schema[1]
    n_norm = len(normalized_shape)
    lead = s.ndim - n_norm
    targets = [list(s.placements)]
    pl = []
    for md, p in enumerate(s.placements):
        if isinstance(p, (Shard, InterleavedShard)) and p.dim < lead:
            pl.append(p)
        elif isinstance(p, Partial) or isinstance(p, RaggedShard) or isinstance(p, (Shard, InterleavedShard)):
            targets[0][md] = Replicate()
            pl.append(Replicate())
        else:
            pl.append(p)
    # weight/bias must be replicate
    rep = tuple(Replicate() for _ in range(s.mesh.ndim))
    n_specs = len(schema.specs)
    input_targets = [tuple(targets[0])] + [rep] * (n_specs - 1)
    stat_shape = tuple(s.shape[:lead]) + tuple(1 for _ in range(n_norm))
    out0 = out_spec(s.mesh, pl, tuple(s.shape), s.dtype)
    stat_dtype = torch.float32 if s.dtype in (torch.float16, torch.bfloat16) else s.dtype
    out1 = out_spec(s.mesh, pl, stat_shape, stat_dtype)
    out2 = out_spec(s.mesh, pl, stat_shape, stat_dtype)
    return OutputSharding([out0, out1, out2], input_targets)


def layer_norm_bwd_rule(schema: OpSchema) -> OutputSharding:
    # native_layer_norm_backward(grad_out, input, normalized_shape, mean,
    #   rstd, weight, bias, output_mask)
    g = schema.specs[0]
    x = schema.specs[1]
    normalized_shape = schema.args_schema[2]
    n_norm = len(normalized_shape)
    lead = x.ndim - n_norm
    mesh = g.mesh
    # grads of input follow input's leading sharding; grads of weight/bias
    # become Partial(sum) on mesh dims where input is sharded on leading dims
    base = []
    partial_wb = []
    for p in x.placements:
        if isinstance(p, (Shard, InterleavedShard)) and p.dim < lead:
            base.append(p)
            partial_wb.append(Partial("sum"))
        elif isinstance(p, Replicate):
            base.append(p)
            partial_wb.append(Replicate())
        else:
            base.append(Replicate())
            partial_wb.append(Replicate())
    rep = tuple(Replicate() for _ in range(mesh.ndim))
    # input targets: grad_out/input/mean/rstd follow base; weight replicate
    n = len(schema.specs)
    input_targets = []
    for i, sp in enumerate(schema.specs):
        if sp.ndim == x.ndim:
            input_targets.append(tuple(base))
        elif sp.ndim == x.ndim and i >= 2:
            input_targets.append(tuple(base))
        else:
            input_targets.append(rep)
    gi = out_spec(mesh, base, tuple(x.shape), x.dtype)
    gw = out_spec(mesh, partial_wb, tuple(normalized_shape), x.dtype)
    gb = out_spec(mesh, partial_wb, tuple(normalized_shape), x.dtype)
    return OutputSharding([gi, gw, gb], input_targets)


def topk_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    k = schema.args_schema[1]
    d = schema.args_schema[2] if len(schema.args_schema) > 2 else -1
    d = d % s.ndim
    targets, pl = [], []
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, (RaggedShard, Partial)):
            targets.append(Replicate())
            pl.append(Replicate())
        else:
            targets.append(p)
            pl.append(p)
    shape = list(s.shape)
    shape[d] = k
    ov = out_spec(s.mesh, pl, shape, s.dtype)
    oi = out_spec(s.mesh, pl, shape, torch.int64)
    return OutputSharding([ov, oi], [tuple(targets)])


def argminmax_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    args = schema.args_schema
    d = args[1] if len(args) > 1 and args[1] is not None else None
    keepdim = args[2] if len(args) > 2 else False
    targets, pl = [], []
    rdims = set(range(s.ndim)) if d is None else {d % s.ndim}
    if keepdim or d is None:
        dim_map = {i: i for i in range(s.ndim)}
    else:
        dim_map = {}
        new = 0
        for i in range(s.ndim):
            if i not in rdims:
                dim_map[i] = new
                new += 1
    for p in s.placements:
        if isinstance(p, (Shard, InterleavedShard)) and p.dim not in rdims and dim_map.get(p.dim) is not None:
            targets.append(p)
            pl.append(Shard(dim_map[p.dim]))
        else:
            targets.append(Replicate())
            pl.append(Replicate())
    shape = _reduced_shape(tuple(s.shape), rdims if d is not None else None, keepdim)
    osp = out_spec(s.mesh, pl, shape, torch.int64)
    return OutputSharding(osp, [tuple(targets)])


def maxmin_dim_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] % s.ndim
    keepdim = schema.args_schema[2] if len(schema.args_schema) > 2 else False
    targets, pl = [], []
    dim_map = {}
    new = 0
    for i in range(s.ndim):
        if i != d:
            dim_map[i] = i if keepdim else new
        new += 0 if i == d and not keepdim else 1
    # simpler: rebuild
    dim_map = {}
    new = 0
    for i in range(s.ndim):
        if i == d:
            if keepdim:
                new += 1
            continue
        dim_map[i] = new
        new += 1
    for p in s.placements:
        if isinstance(p, (Shard, InterleavedShard)) and p.dim != d and dim_map.get(p.dim) is not None:
            targets.append(p)
            pl.append(Shard(dim_map[p.dim]))
        else:
            targets.append(Replicate())
            pl.append(Replicate())
    shape = _reduced_shape(tuple(s.shape), {d}, keepdim)
    ov = out_spec(s.mesh, pl, shape, s.dtype)
    oi = out_spec(s.mesh, pl, shape, torch.int64)
    return OutputSharding([ov, oi], [tuple(targets)])


def cumsum_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] % s.ndim
    targets, pl = [], []
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, (RaggedShard, Partial)):
            targets.append(Replicate())
            pl.append(Replicate())
        else:
            targets.append(p)
            pl.append(p)
    osp = out_spec(s.mesh, pl, tuple(s.shape), s.dtype)
    return OutputSharding(osp, [tuple(targets)])


def prod_rule(schema: OpSchema) -> OutputSharding:
    # prod's Partial combine is a PRODUCT, which our Partial does not model:
    # gather sharded/partial dims instead (breadth op, not a hot path)
    s = schema.specs[0]
    rep = tuple(Replicate() for _ in range(s.mesh.ndim))
    out = _reduction_rule(schema, "sum")
    out.input_targets = [rep]
    sp = out.output_spec
    out.output_spec = out_spec(s.mesh, rep, tuple(sp.shape), sp.dtype)
    return out


def all_rule(schema: OpSchema) -> OutputSharding:
    # logical-AND/OR reduce: same gather-first treatment as prod, but the
    # result dtype is bool
    out = prod_rule(schema)
    sp = out.output_spec
    out.output_spec = out_spec(sp.mesh, sp.placements, tuple(sp.shape), torch.bool)
    return out


def register(dispatcher):
    dispatcher.register_rule(aten.sum, sum_rule)
    dispatcher.register_rule(aten.prod, prod_rule)
    dispatcher.register_rule(aten.all, all_rule)
    dispatcher.register_rule(aten.any, all_rule)
    dispatcher.register_rule(aten.var, prod_rule)  # var over a sharded dim is not Partial-combinable: gather first
    dispatcher.register_rule(aten.count_nonzero, prod_rule)
    dispatcher.register_rule(aten.mean, mean_rule)
    dispatcher.register_rule(aten.amax.default, amax_rule)
    dispatcher.register_rule(aten.amin.default, amin_rule)
    dispatcher.register_rule(aten.max.default, amax_rule)
    dispatcher.register_rule(aten.min.default, amin_rule)
    dispatcher.register_rule(aten.max.dim, maxmin_dim_rule)
    dispatcher.register_rule(aten.min.dim, maxmin_dim_rule)
    dispatcher.register_rule(aten.argmax.default, argminmax_rule)
    dispatcher.register_rule(aten.argmin.default, argminmax_rule)
    dispatcher.register_rule(aten.topk.default, topk_rule)
    dispatcher.register_rule(aten._softmax.default, softmax_rule)
    dispatcher.register_rule(aten._log_softmax.default, softmax_rule)
    dispatcher.register_rule(aten._softmax_backward_data.default, softmax_bwd_rule)
    dispatcher.register_rule(aten._log_softmax_backward_data.default, softmax_bwd_rule)
    dispatcher.register_rule(aten.native_layer_norm.default, layer_norm_rule)
    dispatcher.register_rule(aten.native_layer_norm_backward.default, layer_norm_bwd_rule)
    dispatcher.register_rule(aten.cumsum.default, cumsum_rule)
    if hasattr(aten, "rms_norm"):
        pass  # rms_norm is composite; our models use the fused HIP kernel path
