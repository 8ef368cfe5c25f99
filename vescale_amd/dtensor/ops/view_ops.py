"""View / shape op rules (parity: legacy/vescale/dtensor/ops/view_ops.py +
vescale_view_ops.py, tensor_ops.py slice/cat/stack sections)."""
from __future__ import annotations

from typing import List, Optional, Sequence

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
from .common import out_spec, same_as_input_rule

aten = torch.ops.aten


def _remap(placements, dim_map) -> Optional[List[Placement]]:
    """Remap shard dims through a tensor-dim permutation/mapping.
    dim_map[old_dim] = new_dim (or None if dim disappears)."""
    out: List[Placement] = []
    for p in placements:
        if isinstance(p, InterleavedShard):
            nd = dim_map[p.dim]
            if nd is None:
                return None
            out.append(InterleavedShard(nd, p.interleaved_size))
        elif isinstance(p, Shard):
            nd = dim_map[p.dim]
            if nd is None:
                return None
            out.append(Shard(nd))
        elif isinstance(p, RaggedShard):
            return None  # ragged tensors don't support view ops
        else:
            out.append(p)
    return out


def _norm_dim(d: int, ndim: int) -> int:
    return d % ndim if ndim > 0 else 0


def transpose_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    if schema.op.overloadpacket == aten.t:
        d0, d1 = 0, 1
    else:
        d0 = _norm_dim(schema.args_schema[1], s.ndim)
        d1 = _norm_dim(schema.args_schema[2], s.ndim)
    dim_map = list(range(s.ndim))
    dim_map[d0], dim_map[d1] = d1, d0
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = list(s.shape)
    shape[d0], shape[d1] = shape[d1], shape[d0]
    return OutputSharding(out_spec(s.mesh, pl, shape, s.dtype), None)


def permute_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    perm = [_norm_dim(d, s.ndim) for d in schema.args_schema[1]]
    # dim_map: old dim -> new position
    dim_map = [0] * s.ndim
    for newpos, old in enumerate(perm):
        dim_map[old] = newpos
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = [s.shape[d] for d in perm]
    return OutputSharding(out_spec(s.mesh, pl, shape, s.dtype), None)


def _view_dim_map(in_shape: Sequence[int], out_shape: Sequence[int]) -> Optional[dict]:
    """Map input dims -> output dims when the reshape splits/merges
    contiguous groups.  Returns {in_dim: (out_dim, leading)} where leading
    means the in dim is the OUTERMOST factor of that output group (required
    for sharding to survive a merge) — or the in dim maps to the outermost
    out dim of a split."""
    # group boundaries by equal prefix products
    in_prod = [1]
    for s in in_shape:
        in_prod.append(in_prod[-1] * s)
    out_prod = [1]
    for s in out_shape:
        out_prod.append(out_prod[-1] * s)
    common = sorted(set(in_prod) & set(out_prod))
    mapping = {}
    for lo, hi in zip(common[:-1], common[1:]):
        # input dims covering (lo, hi]
        in_dims = [i for i in range(len(in_shape)) if in_prod[i] >= lo and in_prod[i + 1] <= hi and in_prod[i] < in_prod[i + 1]]
        out_dims = [i for i in range(len(out_shape)) if out_prod[i] >= lo and out_prod[i + 1] <= hi and out_prod[i] < out_prod[i + 1]]
        if in_dims and out_dims:
            # outermost input dim of the group maps to outermost output dim
            mapping[in_dims[0]] = out_dims[0]
    return mapping


def view_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    target = list(schema.args_schema[1])
    # resolve -1
    numel = 1
    for x in s.shape:
        numel *= x
    known = 1
    neg = -1
    for i, t in enumerate(target):
        if t == -1:
            neg = i
        else:
            known *= t
    if neg >= 0:
        target[neg] = numel // max(1, known)

    if all(p.is_replicate() or p.is_partial() for p in s.placements):
        return OutputSharding(out_spec(s.mesh, s.placements, target, s.dtype), None)

    mapping = _view_dim_map(tuple(s.shape), tuple(target))
    new_placements: List[Placement] = []
    ok = True
    for p in s.placements:
        if isinstance(p, Shard):
            nd = mapping.get(p.dim)
            if nd is None:
                ok = False
                break
            # sharding survives only if local view works: local size on the
            # group must still factor.  Even sharding of the outermost factor
            # always works; uneven merge does not.
            new_placements.append(Shard(nd))
        elif isinstance(p, (InterleavedShard, RaggedShard)):
            ok = False
            break
        else:
            new_placements.append(p)
    if not ok:
        return None  # dispatcher falls back to replicate

    # local shape check: compute local target shape by dividing the sharded
    # out dims — require divisibility, else fall back
    mesh = s.mesh
    coord = mesh.get_coordinate()
    return OutputSharding(out_spec(mesh, new_placements, target, s.dtype), None)


def expand_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    target = list(schema.args_schema[1])
    offset = len(target) - s.ndim
    for i, t in enumerate(target):
        if t == -1:
            target[i] = s.shape[i - offset]
    dim_map = {d: d + offset for d in range(s.ndim)}
    pl: List[Placement] = []
    for p in s.placements:
        if isinstance(p, Shard):
            pl.append(Shard(dim_map[p.dim]))
        elif isinstance(p, (InterleavedShard, RaggedShard)):
            return None
        else:
            pl.append(p)
    return OutputSharding(out_spec(s.mesh, pl, target, s.dtype), None)


def unsqueeze_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1]
    d = d % (s.ndim + 1)
    dim_map = [i if i < d else i + 1 for i in range(s.ndim)]
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = list(s.shape)
    shape.insert(d, 1)
    return OutputSharding(out_spec(s.mesh, pl, shape, s.dtype), None)


def squeeze_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    if len(schema.args_schema) > 1:
        ds = schema.args_schema[1]
        if isinstance(ds, int):
            ds = [ds]
        ds = [_norm_dim(d, s.ndim) for d in ds]
    else:
        ds = [i for i, x in enumerate(s.shape) if x == 1]
    ds = [d for d in ds if s.shape[d] == 1]
    dim_map: List[Optional[int]] = []
    new = 0
    for i in range(s.ndim):
        if i in ds:
            dim_map.append(None)
        else:
            dim_map.append(new)
            new += 1
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = [x for i, x in enumerate(s.shape) if i not in ds]
    return OutputSharding(out_spec(s.mesh, pl, shape, s.dtype), None)


def slice_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = _norm_dim(schema.args_schema[1] if len(schema.args_schema) > 1 else 0, s.ndim)
    start = schema.args_schema[2] if len(schema.args_schema) > 2 else 0
    end = schema.args_schema[3] if len(schema.args_schema) > 3 else None
    step = schema.args_schema[4] if len(schema.args_schema) > 4 else 1
    size = s.shape[d]
    start = 0 if start is None else (start + size if start < 0 else start)
    end = size if end is None else min(end + size if end < 0 else end, size)
    start = min(start, size)
    length = max(0, -(-(end - start) // step))
    # full-range slice is a no-op
    if start == 0 and end >= size and step == 1:
        return same_as_input_rule(schema)
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard):
            return None  # slicing the sharded dim -> fallback
    shape = list(s.shape)
    shape[d] = length
    return OutputSharding(out_spec(s.mesh, s.placements, shape, s.dtype), None)


def select_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = _norm_dim(schema.args_schema[1], s.ndim)
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard):
            return None
    dim_map = [i if i < d else (None if i == d else i - 1) for i in range(s.ndim)]
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = [x for i, x in enumerate(s.shape) if i != d]
    return OutputSharding(out_spec(s.mesh, pl, shape, s.dtype), None)


def cat_rule(schema: OpSchema) -> OutputSharding:
    tensors = schema.args_schema[0]
    specs = [t for t in tensors if isinstance(t, DTensorSpec)]
    if not specs:
        return None
    d = schema.args_schema[1] if len(schema.args_schema) > 1 else 0
    d = _norm_dim(d, specs[0].ndim)
    mesh = specs[0].mesh
    # special case: cat of equal-size Shard(d) chunks ON the cat dim is the
    # inverse of the interleaved split -> output InterleavedShard(d, n)
    # (packed-QKV backward path; no communication)
    def shard_on_d(sp, md):
        p = sp.placements[md]
        return isinstance(p, Shard) and not isinstance(p, InterleavedShard) and p.dim == d

    same_size = all(tuple(sp.shape) == tuple(specs[0].shape) for sp in specs)
    il_base: List[Placement] = []
    ok_il = same_size and len(specs) > 1
    if ok_il:
        for md in range(mesh.ndim):
            if all(shard_on_d(sp, md) for sp in specs):
                il_base.append(InterleavedShard(d, len(specs)))
            elif all(sp.placements[md] == specs[0].placements[md] and not sp.placements[md].is_partial() and not (isinstance(sp.placements[md], (Shard, InterleavedShard)) and sp.placements[md].dim == d) for sp in specs):
                il_base.append(specs[0].placements[md])
            else:
                ok_il = False
                break
    if ok_il and any(isinstance(p, InterleavedShard) and p.dim == d for p in il_base):
        shape = list(specs[0].shape)
        shape[d] = sum(sp.shape[d] for sp in specs)
        return OutputSharding(out_spec(mesh, il_base, shape, specs[0].dtype), None)

    # general case: all inputs align to the first's placements; cat dim must
    # not be sharded
    base = []
    for p in specs[0].placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard) or isinstance(p, Partial):
            base.append(Replicate())
        else:
            base.append(p)
    base = tuple(base)
    targets = [base for _ in specs]
    shape = list(specs[0].shape)
    shape[d] = sum(sp.shape[d] for sp in specs)
    return OutputSharding(out_spec(mesh, base, shape, specs[0].dtype), targets)


def stack_rule(schema: OpSchema) -> OutputSharding:
    tensors = schema.args_schema[0]
    specs = [t for t in tensors if isinstance(t, DTensorSpec)]
    if not specs:
        return None
    d = schema.args_schema[1] if len(schema.args_schema) > 1 else 0
    nd = specs[0].ndim + 1
    d = _norm_dim(d, nd)
    mesh = specs[0].mesh
    base: List[Placement] = []
    for p in specs[0].placements:
        if isinstance(p, (InterleavedShard, RaggedShard)) or isinstance(p, Partial):
            base.append(Replicate())
        elif isinstance(p, Shard):
            base.append(Shard(p.dim))
        else:
            base.append(p)
    targets = [tuple(base) for _ in specs]
    out_pl = []
    for p in base:
        if isinstance(p, Shard):
            out_pl.append(Shard(p.dim if p.dim < d else p.dim + 1))
        else:
            out_pl.append(p)
    shape = list(specs[0].shape)
    shape.insert(d, len(tensors))
    return OutputSharding(out_spec(mesh, out_pl, shape, specs[0].dtype), targets)


def split_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    split = schema.args_schema[1]
    d = schema.args_schema[2] if len(schema.args_schema) > 2 else 0
    d = _norm_dim(d, s.ndim)
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard):
            return None
    size = s.shape[d]
    if isinstance(split, int):
        sizes = [min(split, size - i * split) for i in range((size + split - 1) // split)]
    else:
        sizes = list(split)
    outs = []
    for sz in sizes:
        shape = list(s.shape)
        shape[d] = sz
        outs.append(out_spec(s.mesh, s.placements, shape, s.dtype))
    return OutputSharding(outs, None)


def unbind_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    d = schema.args_schema[1] if len(schema.args_schema) > 1 else 0
    d = _norm_dim(d, s.ndim)
    for p in s.placements:
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard):
            return None
    dim_map = [i if i < d else (None if i == d else i - 1) for i in range(s.ndim)]
    pl = _remap(s.placements, dim_map)
    if pl is None:
        return None
    shape = [x for i, x in enumerate(s.shape) if i != d]
    osp = out_spec(s.mesh, pl, shape, s.dtype)
    return OutputSharding([osp] * s.shape[d], None)


def _handler_view(dispatcher, op, args, kwargs):
    """view/reshape/_unsafe_view with LOCALIZED shape args (the reference's
    _adjust_shape_and_stride_args, vescale/dtensor/_sharding_prop.py:228):
    the target shape the user wrote is GLOBAL; the local op needs each
    sharded output dim divided by its shard count."""
    from ..dtensor import DTensor

    x = args[0]
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    mesh = spec.mesh
    target = list(args[1])
    numel = 1
    for s in spec.shape:
        numel *= s
    known, neg = 1, -1
    for i, t in enumerate(target):
        if t == -1:
            neg = i
        else:
            known *= t
    if neg >= 0:
        target[neg] = numel // max(1, known)

    mapping = _view_dim_map(tuple(spec.shape), tuple(target))

    out_placements: List[Placement] = []
    bad_dims = []
    for md, p in enumerate(spec.placements):
        if isinstance(p, InterleavedShard):
            nd = mapping.get(p.dim)
            if nd is None or target[nd] % (p.interleaved_size * mesh.size(md)) != 0 or spec.shape[p.dim] != target[nd]:
                bad_dims.append(md)
                out_placements.append(Replicate())
            else:
                out_placements.append(InterleavedShard(nd, p.interleaved_size))
        elif isinstance(p, Shard):
            nd = mapping.get(p.dim)
            g_in = spec.shape[p.dim]
            l_in = x._local_tensor.shape[p.dim]
            # local out size = target[nd] * l_in / g_in must be integral
            if nd is None or g_in == 0 or (target[nd] * l_in) % g_in != 0:
                bad_dims.append(md)
                out_placements.append(Replicate())
            else:
                out_placements.append(Shard(nd))
        elif isinstance(p, RaggedShard):
            bad_dims.append(md)
            out_placements.append(Replicate())
        else:
            out_placements.append(p)

    if bad_dims:
        placements = [
            Replicate() if md in bad_dims else p
            for md, p in enumerate(spec.placements)
        ]
        x = x.redistribute(placements=placements)
        spec = x._spec

    # localize target shape via the input's actual local sizes
    inv_mapping = {}
    for md, p in enumerate(spec.placements):
        if isinstance(p, (Shard, InterleavedShard)):
            nd = mapping.get(p.dim)
            if nd is not None:
                inv_mapping[nd] = p.dim
    local_target = list(target)
    for md, p in enumerate(out_placements):
        if isinstance(p, (Shard, InterleavedShard)):
            d_in = inv_mapping.get(p.dim)
            if d_in is None:
                continue
            g_in = spec.shape[d_in]
            l_in = x._local_tensor.shape[d_in]
            local_target[p.dim] = target[p.dim] * l_in // g_in
    try:
        local = op(x._local_tensor, local_target, *args[2:], **kwargs)
    except RuntimeError:
        # view on a non-contiguous local (e.g. HF attention's
        # transpose().reshape() path): reshape semantics — copy when the
        # strides don't permit a view
        local = x._local_tensor.reshape(local_target)
    osp = out_spec(mesh, out_placements, target, spec.dtype)
    return DTensor(local, osp, requires_grad=local.requires_grad)


def _handler_expand(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    x = args[0]
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    mesh = spec.mesh
    target = list(args[1])
    offset = len(target) - spec.ndim
    for i, t in enumerate(target):
        if t == -1:
            target[i] = spec.shape[i - offset]
    out_placements = []
    bad = []
    for md, p in enumerate(spec.placements):
        if isinstance(p, Shard):
            out_placements.append(Shard(p.dim + offset))
        elif isinstance(p, (InterleavedShard, RaggedShard)):
            bad.append(md)
            out_placements.append(Replicate())
        else:
            out_placements.append(p)
    if bad:
        x = x.redistribute(placements=[
            Replicate() if md in bad else p for md, p in enumerate(spec.placements)
        ])
    local_target = list(target)
    for md, p in enumerate(out_placements):
        if isinstance(p, Shard):
            local_target[p.dim] = x._local_tensor.shape[p.dim - offset]
    local = op(x._local_tensor, local_target, *args[2:], **kwargs)
    osp = out_spec(mesh, out_placements, target, spec.dtype)
    return DTensor(local, osp, requires_grad=local.requires_grad)


def _handler_split(dispatcher, op, args, kwargs):
    """split on a DTensor, InterleavedShard-aware (the packed-QKV TP path:
    splitting an IS(d, y) tensor into its y sections needs the LOCAL split
    size rewritten to local_dim/y and yields Shard(d) sections)."""
    from ..dtensor import DTensor
    from ..redistribute import redistribute_local_tensor

    x = args[0]
    if not isinstance(x, DTensor):
        return op(*args, **kwargs)
    spec = x._spec
    mesh = spec.mesh
    split = args[1]
    d = args[2] if len(args) > 2 else kwargs.get("dim", 0)
    d = _norm_dim(d, spec.ndim)
    sizes = (
        [split] * ((spec.shape[d] + split - 1) // split)
        if isinstance(split, int)
        else list(split)
    )
    if isinstance(split, int):
        sizes = [min(split, spec.shape[d] - i * split) for i in range(len(sizes))]
    n_sec = len(sizes)

    # find an IS placement on dim d with matching section structure
    is_md = None
    for md, p in enumerate(spec.placements):
        if isinstance(p, InterleavedShard) and p.dim == d and p.interleaved_size == n_sec and all(s == sizes[0] for s in sizes):
            is_md = md
            break
    local = x._local_tensor
    if is_md is not None:
        # other mesh dims must not shard d
        bad = [
            md for md, p in enumerate(spec.placements)
            if md != is_md and ((isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard))
        ]
        if not bad:
            local_sz = local.shape[d] // n_sec
            chunks = torch.split(local, local_sz, dim=d)
            outs = []
            for i, c in enumerate(chunks):
                pl = tuple(
                    Shard(d) if md == is_md else p
                    for md, p in enumerate(spec.placements)
                )
                shape = list(spec.shape)
                shape[d] = sizes[i]
                outs.append(
                    DTensor(
                        c,
                        out_spec(mesh, pl, shape, spec.dtype),
                        requires_grad=c.requires_grad,
                    )
                )
            return outs

    # general path: dim d must be unsharded; redistribute if needed
    needs = [
        md for md, p in enumerate(spec.placements)
        if (isinstance(p, (Shard, InterleavedShard)) and p.dim == d) or isinstance(p, RaggedShard)
    ]
    placements = list(spec.placements)
    if needs:
        for md in needs:
            placements[md] = Replicate()
        x = x.redistribute(placements=placements)
        spec = x._spec
        local = x._local_tensor
    chunks = torch.split(local, split if isinstance(split, int) else sizes, dim=d)
    outs = []
    for i, c in enumerate(chunks):
        shape = list(spec.shape)
        shape[d] = sizes[i]
        outs.append(
            DTensor(c, out_spec(mesh, spec.placements, shape, spec.dtype),
                    requires_grad=c.requires_grad)
        )
    return outs


def register(dispatcher):
    for op in (aten.detach, aten.alias, aten.clone, aten.contiguous, aten._unsafe_view_copy if hasattr(aten, "_unsafe_view_copy") else aten.alias):
        dispatcher.register_rule(op, same_as_input_rule)
    dispatcher.register_rule(aten.t.default, transpose_rule)
    dispatcher.register_rule(aten.transpose.int, transpose_rule)
    dispatcher.register_rule(aten.permute.default, permute_rule)
    dispatcher.register_handler(aten.view.default, _handler_view)
    dispatcher.register_handler(aten._unsafe_view.default, _handler_view)
    dispatcher.register_handler(aten.reshape.default, _handler_view)
    dispatcher.register_handler(aten.expand.default, _handler_expand)
    dispatcher.register_rule(aten.unsqueeze.default, unsqueeze_rule)
    dispatcher.register_rule(aten.squeeze, squeeze_rule)
    dispatcher.register_rule(aten.slice.Tensor, slice_rule)
    dispatcher.register_rule(aten.select.int, select_rule)
    dispatcher.register_rule(aten.cat.default, cat_rule)
    dispatcher.register_rule(aten.stack.default, stack_rule)
    dispatcher.register_handler(aten.split.Tensor, _handler_split)
    dispatcher.register_handler(aten.split_with_sizes.default, _handler_split)
    dispatcher.register_rule(aten.unbind.int, unbind_rule)
