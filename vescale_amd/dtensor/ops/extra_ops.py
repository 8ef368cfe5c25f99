"""Extra aten coverage: the reference's "enabled DTensor ops" README list
(legacy/vescale/dtensor/README.md:56-74) beyond what the core rule tables
already handle — sort / bucketize / searchsorted as declarative rules, and
one_hot / index_put(_) / index_add(_) / _unique2 / expand_as as eager
handlers (reference implements these as dispatch pre-patches,
legacy/vescale/dtensor/_dispatch_patch.py:62-133; here they are ordinary
registry handlers on the same dispatcher).

Conservative semantics: ops whose result mixes element positions across the
sharded dim (sort on the shard dim, index scatter/add with arbitrary
indices, unique) gather to Replicate first; element-local ops (bucketize,
one_hot) keep the input's sharding.
"""
from __future__ import annotations

import torch

from .._dtensor_spec import DTensorSpec
from .._op_schema import OpSchema, OutputSharding
from ..placement_types import (
    InterleavedShard,
    Partial,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
    _StridedRaggedShard,
)
from .common import contiguous_stride, out_spec

aten = torch.ops.aten


# ---------------------------------------------------------------------------
# declarative rules
# ---------------------------------------------------------------------------
def sort_rule(schema: OpSchema) -> OutputSharding:
    """sort along dim: any placement sharding the sort dim (or position-
    ambiguous Ragged/Partial) is replicated; other shard dims propagate.
    Outputs (values, indices) share the input's (possibly adjusted)
    placements; indices are int64."""
    s = schema.specs[0]
    d = -1
    if len(schema.args_schema) > 1 and isinstance(schema.args_schema[1], int):
        d = schema.args_schema[1]
    d = schema.kwargs_schema.get("dim", d) % s.ndim
    targets, pl = [], []
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(
            p, (RaggedShard, Partial)
        ):
            targets.append(Replicate())
            pl.append(Replicate())
        else:
            targets.append(p)
            pl.append(p)
    ov = out_spec(s.mesh, pl, tuple(s.shape), s.dtype)
    oi = out_spec(s.mesh, pl, tuple(s.shape), torch.int64)
    return OutputSharding([ov, oi], [tuple(targets)])


def bucketize_rule(schema: OpSchema) -> OutputSharding:
    """bucketize(input, boundaries): element-local in `input`, so its shard
    placements propagate (Partial would change values -> replicate);
    `boundaries` must be whole on every rank."""
    s = schema.specs[0]
    pl = [Replicate() if isinstance(p, Partial) else p for p in s.placements]
    targets = [tuple(pl)]
    if len(schema.specs) > 1:
        targets.append(tuple(Replicate() for _ in range(s.mesh.ndim)))
    out_dtype = (
        torch.int32 if schema.kwargs_schema.get("out_int32", False) else torch.int64
    )
    osp = out_spec(s.mesh, pl, tuple(s.shape), out_dtype)
    return OutputSharding(osp, targets)


def searchsorted_rule(schema: OpSchema) -> OutputSharding:
    """searchsorted(sorted_sequence, input): the sorted sequence must be
    whole; the probed values are element-local and keep their sharding."""
    seq = schema.specs[0]
    val = schema.specs[1] if len(schema.specs) > 1 else None
    rep = tuple(Replicate() for _ in range(seq.mesh.ndim))
    if val is None:  # scalar probe
        osp = out_spec(seq.mesh, rep, (), torch.int64)
        return OutputSharding(osp, [rep])
    pl = [Replicate() if isinstance(p, Partial) else p for p in val.placements]
    out_dtype = (
        torch.int32 if schema.kwargs_schema.get("out_int32", False) else torch.int64
    )
    osp = out_spec(val.mesh, pl, tuple(val.shape), out_dtype)
    return OutputSharding(osp, [rep, tuple(pl)])


# ---------------------------------------------------------------------------
# eager handlers
# ---------------------------------------------------------------------------
def _replicate_local(x):
    """DTensor -> its full (replicated) local tensor; passthrough otherwise."""
    from ..dtensor import DTensor

    if isinstance(x, DTensor):
        rep = tuple(Replicate() for _ in range(x._spec.mesh.ndim))
        return x.redistribute(placements=rep)._local_tensor
    if isinstance(x, (list, tuple)):
        return type(x)(_replicate_local(y) for y in x)
    return x


def _wrap_replicate(t: torch.Tensor, mesh):
    from ..dtensor import DTensor

    tm = TensorMeta(t.shape, contiguous_stride(t.shape), t.dtype)
    sp = DTensorSpec(mesh, tuple(Replicate() for _ in range(mesh.ndim)), tm)
    return DTensor(t, sp, requires_grad=t.requires_grad)


def _first_mesh(args, kwargs):
    from ..dtensor import DTensor

    def walk(x):
        if isinstance(x, DTensor):
            return x._spec.mesh
        if isinstance(x, (list, tuple)):
            for y in x:
                m = walk(y)
                if m is not None:
                    return m
        return None

    for a in args:
        m = walk(a)
        if m is not None:
            return m
    for a in kwargs.values():
        m = walk(a)
        if m is not None:
            return m
    raise RuntimeError("no DTensor args")


def _handler_replicate_compute(dispatcher, op, args, kwargs):
    """Gather every DTensor arg to Replicate, run the op locally, and wrap
    tensor outputs as Replicate DTensors.  Correct for any op; used for the
    position-scrambling tail ops (unique, expand_as) where a sharded
    fast path has no stable meaning."""
    mesh = _first_mesh(args, kwargs)
    local_args = tuple(_replicate_local(a) for a in args)
    local_kwargs = {k: _replicate_local(v) for k, v in kwargs.items()}
    res = op(*local_args, **local_kwargs)
    if isinstance(res, torch.Tensor):
        return _wrap_replicate(res, mesh)
    if isinstance(res, (list, tuple)):
        return type(res)(
            _wrap_replicate(r, mesh) if isinstance(r, torch.Tensor) else r for r in res
        )
    return res


def _writeback_inplace(self_dt, full_result: torch.Tensor):
    """Scatter a full-tensor result back into an in-place target's local
    shard: wrap as Replicate, redistribute to the target's placements, and
    copy into its local tensor."""
    from ..dtensor import DTensor  # noqa: F401

    rep_dt = _wrap_replicate(full_result, self_dt._spec.mesh)
    shard_dt = rep_dt.redistribute(placements=self_dt._spec.placements)
    self_dt._local_tensor.copy_(shard_dt._local_tensor)
    return self_dt


def _handler_index_write(dispatcher, op, args, kwargs):
    """index_put(_), index_add(_): indices address GLOBAL positions, so the
    write is performed on the gathered tensor and (for the in-place
    variants) scattered back into the caller's shard.  Reference treats
    these as dispatch pre-patches (_dispatch_patch.py:62-133)."""
    from ..dtensor import DTensor

    self_dt = args[0]
    inplace = op._schema.name.endswith("_")
    full_self = _replicate_local(self_dt)
    if isinstance(self_dt, DTensor):
        full_self = full_self.clone()
    local_rest = tuple(_replicate_local(a) for a in args[1:])
    local_kwargs = {k: _replicate_local(v) for k, v in kwargs.items()}
    res = op(full_self, *local_rest, **local_kwargs)
    out = res if isinstance(res, torch.Tensor) else full_self
    if inplace and isinstance(self_dt, DTensor):
        return _writeback_inplace(self_dt, out)
    if inplace:
        # plain self with DTensor index/source args: op already mutated
        # full_self (== self); in-place contract returns self itself
        return self_dt
    mesh = self_dt._spec.mesh if isinstance(self_dt, DTensor) else _first_mesh(args, kwargs)
    return _wrap_replicate(out, mesh)


def _handler_one_hot(dispatcher, op, args, kwargs):
    """one_hot is element-local: each index row expands into a new trailing
    dim, so the input's shard placements carry over unchanged.  num_classes
    must be GLOBAL: when defaulted (-1), the class count is the mesh-wide
    max index + 1 (a local max would give ranks different widths)."""
    from ..dtensor import DTensor
    from .. import _collective_utils as cc

    x = args[0]
    num_classes = args[1] if len(args) > 1 else kwargs.get("num_classes", -1)
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    local = x._local_tensor
    if num_classes is None or num_classes < 0:
        mx = local.max().reshape(1).clone() if local.numel() else torch.zeros(
            1, dtype=torch.int64, device=local.device
        )
        for md in range(spec.mesh.ndim):
            cc.mesh_all_reduce(mx, spec.mesh, "max", md)
        num_classes = int(mx.item()) + 1
    res = op(local, num_classes)
    pl = [Replicate() if isinstance(p, Partial) else p for p in spec.placements]
    shape = tuple(spec.shape) + (num_classes,)
    tm = TensorMeta(torch.Size(shape), contiguous_stride(shape), res.dtype)
    sp = DTensorSpec(spec.mesh, tuple(pl), tm)
    return DTensor(res, sp, requires_grad=False)


def new_factory_rule(schema: OpSchema) -> OutputSharding:
    """new_zeros/new_ones/new_full/new_empty(_strided): fresh tensor of an
    explicit GLOBAL shape on the input's mesh — Replicate output (the new
    shape has no relation to the input's sharding)."""
    s = schema.specs[0]
    shape = tuple(schema.args_schema[1]) if len(schema.args_schema) > 1 else ()
    dtype = schema.kwargs_schema.get("dtype") or s.dtype
    rep = [Replicate()] * s.mesh.ndim
    return OutputSharding(out_spec(s.mesh, rep, shape, dtype), None)


def _handler_linear(dispatcher, op, args, kwargs):
    """aten.linear on DTensors: re-dispatch as x @ w.T (+ b) so the matmul
    rule tables drive sharding (colwise/rowwise TP propagate)."""
    x, w = args[0], args[1]
    b = args[2] if len(args) > 2 else None
    out = x @ w.t()
    if b is not None:
        out = out + b
    return out


def _handler_trilu(dispatcher, op, args, kwargs):
    """tril/triu are POSITION-dependent, not pointwise (caught by the op
    parity sweep: plain local tril on a Shard(0) input masked as if local
    row 0 were global row 0).  A shard of the last two dims still computes
    LOCALLY by shifting the diagonal by the shard's global offset:
    row-shard r0: keep col <= (r0+i) + d  ==  local diagonal d + r0;
    col-shard c0: keep (c0+j) - i <= d    ==  local diagonal d - c0.
    Other layered placements gather first."""
    from ..dtensor import DTensor

    x = args[0]
    d = args[1] if len(args) > 1 else kwargs.get("diagonal", 0)
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    nd = len(spec.shape)
    row_dim, col_dim = nd - 2, nd - 1
    shift = 0
    ok_local = True
    for md, p in enumerate(spec.placements):
        if isinstance(p, _StridedRaggedShard) or isinstance(p, (RaggedShard, InterleavedShard)):
            ok_local = False
        elif isinstance(p, Shard):
            w = spec.mesh.size(md)
            my = spec.mesh.get_coordinate()[md]
            if p.dim % nd == row_dim:
                shift += Shard.chunk_offset(spec.shape[row_dim], w, my)
            elif p.dim % nd == col_dim:
                shift -= Shard.chunk_offset(spec.shape[col_dim], w, my)
    if not ok_local:
        return _handler_replicate_compute(dispatcher, op, args, kwargs)
    res = op(x._local_tensor, d + shift)
    tm = TensorMeta(spec.shape, spec.tensor_meta.stride, res.dtype)
    out = DTensor(res, DTensorSpec(spec.mesh, spec.placements, tm),
                  requires_grad=res.requires_grad)
    if op._schema.name.endswith("_"):
        x._local_tensor.copy_(res)
        return x
    return out


def register(dispatcher):
    for ov in (aten.sort.default, aten.sort.stable):
        dispatcher.register_rule(ov, sort_rule)
    dispatcher.register_rule(aten.bucketize.Tensor, bucketize_rule)
    dispatcher.register_rule(aten.searchsorted.Tensor, searchsorted_rule)

    dispatcher.register_handler(aten.one_hot.default, _handler_one_hot)
    for ov in (
        aten.index_put.default,
        aten.index_put_.default,
        aten.index_put.hacked_twin,
        aten.index_put_.hacked_twin,
        aten.index_add.default,
        aten.index_add_.default,
    ):
        dispatcher.register_handler(ov, _handler_index_write)
    for ov in (aten.tril.default, aten.triu.default, aten.tril_.default,
               aten.triu_.default):
        dispatcher.register_handler(ov, _handler_trilu)
    dispatcher.register_handler(aten._unique2.default, _handler_replicate_compute)
    dispatcher.register_handler(aten.expand_as.default, _handler_replicate_compute)

    # breadth sweep ---------------------------------
    for ov in (aten.new_zeros.default, aten.new_ones.default,
               aten.new_full.default, aten.new_empty.default,
               aten.new_empty_strided.default):
        dispatcher.register_rule(ov, new_factory_rule)
    # in-place scatter with GLOBAL indices: gather-apply-writeback like
    # index_put_ (scatter.src/value out-of-place rules live in tensor_ops)
    for ov in (aten.scatter_.src, aten.scatter_.value):
        dispatcher.register_handler(ov, _handler_index_write)
    dispatcher.register_handler(aten.linear.default, _handler_linear)
    # data-dependent shapes / conservative tail ops: gather to Replicate
    for ov in (
        aten.nonzero.default,
        aten.repeat.default,
        aten.constant_pad_nd.default,
        aten.slice_backward.default,
        aten.select_backward.default,
        aten.slice_scatter.default,
        aten.mse_loss.default,
        aten.mse_loss_backward.default,
        aten.nll_loss2d_forward.default,
        aten.nll_loss2d_backward.default,
        aten.embedding_renorm_.default,
        aten.argsort.default,
    ):
        dispatcher.register_handler(ov, _handler_replicate_compute)
