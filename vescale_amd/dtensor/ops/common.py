"""Shared rule helpers: broadcast mapping, pointwise rule, spec builders.

Parity concept: legacy/vescale/dtensor/ops/common_rules.py (einop/pointwise)
re-designed around explicit per-mesh-dim placement voting.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

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
    TensorMeta,
)


def contiguous_stride(shape: Sequence[int]) -> Tuple[int, ...]:
    st, acc = [], 1
    for s in reversed(list(shape)):
        st.append(acc)
        acc *= s
    return tuple(reversed(st))


def out_spec(mesh, placements, shape, dtype) -> DTensorSpec:
    tm = TensorMeta(torch.Size(shape), contiguous_stride(shape), dtype)
    return DTensorSpec(mesh, tuple(placements), tm)


def bcast_dim_map(in_shape: Sequence[int], out_shape: Sequence[int]) -> List[Optional[int]]:
    """For each output dim, the corresponding input dim or None if the input
    broadcasts there (missing dim or size-1 vs >1)."""
    offset = len(out_shape) - len(in_shape)
    m: List[Optional[int]] = []
    for od in range(len(out_shape)):
        idx = od - offset
        if idx < 0:
            m.append(None)
        elif in_shape[idx] == out_shape[od]:
            m.append(idx)
        else:  # size-1 broadcast
            m.append(None)
    return m


# ops where Partial(sum) inputs may pass through untouched when EVERY tensor
# input on that mesh dim is Partial(sum) (linearity: sum_i(a_i + b_i) = A+B)
LINEAR_ADD_OPS = {
    "add", "add_", "sub", "sub_", "neg", "neg_", "sum",
}
# unary/scale ops where a single Partial input passes through (f(sum) = sum f
# for scaling by non-tensor scalar)
LINEAR_SCALE_OPS = {
    "mul", "mul_", "div", "div_", "neg", "neg_", "detach", "clone", "_to_copy",
}


def _op_base_name(op) -> str:
    return op.overloadpacket.__name__.rstrip("_") + ("_" if op.overloadpacket.__name__.endswith("_") else "")


def pointwise_rule(schema: OpSchema, *, inplace: bool = False, out_dtype=None) -> OutputSharding:
    """Generic elementwise rule with broadcasting.

    Decides output placements per mesh dim and input redistribution targets.
    """
    op = schema.op
    specs = schema.specs
    mesh = specs[0].mesh
    name = op.overloadpacket.__name__
    base = name.rstrip("_")
    is_inplace = name.endswith("_") or inplace

    shapes = [tuple(s.shape) for s in specs]
    try:
        out_shape = torch.broadcast_shapes(*shapes)
    except RuntimeError:
        out_shape = shapes[0]
    if is_inplace:
        out_shape = shapes[0]
    maps = [bcast_dim_map(sh, out_shape) for sh in shapes]

    n_tensor_scalar_inputs = len(specs)
    out_placements: List[Placement] = []
    targets: List[List[Placement]] = [list(s.placements) for s in specs]

    for md in range(mesh.ndim):
        ps = [s.placements[md] for s in specs]
        # pick sharding winner: inplace -> arg0 dictates; else first sharded
        winner: Optional[Placement] = None
        if is_inplace:
            winner = ps[0]
        else:
            for i, p in enumerate(ps):
                if isinstance(p, (Shard, InterleavedShard)):
                    # map to output dim
                    od = _map_to_out(maps[i], p.dim)
                    if od is None:
                        continue
                    winner = (
                        InterleavedShard(od, p.interleaved_size)
                        if isinstance(p, InterleavedShard)
                        else Shard(od)
                    )
                    break
                if isinstance(p, RaggedShard):
                    winner = p
                    break
            if winner is None:
                # partial handling
                partials = [p for p in ps if isinstance(p, Partial)]
                if partials:
                    all_partial = all(isinstance(p, Partial) for p in ps)
                    if all_partial and base in LINEAR_ADD_OPS:
                        winner = partials[0]
                    elif len(specs) == 1 and base in (LINEAR_SCALE_OPS | LINEAR_ADD_OPS):
                        winner = partials[0]
                    elif len(partials) == 1 and len(specs) == 1:
                        winner = Replicate()
                    else:
                        winner = Replicate()
                else:
                    winner = Replicate()

        # now set targets for each input on this mesh dim
        for i, (p, m) in enumerate(zip(ps, maps)):
            if isinstance(winner, (Shard, InterleavedShard)):
                od = winner.dim
                idx = m[od]
                if idx is None:
                    # this input broadcasts on the sharded dim -> Replicate
                    targets[i][md] = _deparallelize(p)
                else:
                    want = (
                        InterleavedShard(idx, winner.interleaved_size)
                        if isinstance(winner, InterleavedShard)
                        else Shard(idx)
                    )
                    targets[i][md] = want
            elif isinstance(winner, RaggedShard):
                targets[i][md] = winner
            elif isinstance(winner, Partial):
                targets[i][md] = p if isinstance(p, Partial) else p
            else:
                targets[i][md] = _deparallelize(p)
        out_placements.append(winner)

    dtype = out_dtype or specs[0].dtype
    # honor explicit dtype kwargs (e.g. _to_copy)
    kd = schema.kwargs_schema.get("dtype")
    if isinstance(kd, torch.dtype):
        dtype = kd
    osp = out_spec(mesh, out_placements, out_shape, dtype)
    return OutputSharding(output_spec=osp, input_targets=[tuple(t) for t in targets])


def _map_to_out(m: List[Optional[int]], in_dim: int) -> Optional[int]:
    for od, idx in enumerate(m):
        if idx == in_dim:
            return od
    return None


def _deparallelize(p: Placement) -> Placement:
    """What a non-winning input must become: Partial -> Replicate (cannot
    combine partial values elementwise), Shard stays only if it equals the
    winner (handled by caller)."""
    if isinstance(p, Partial):
        return Replicate()
    if isinstance(p, (Shard, InterleavedShard, RaggedShard)):
        return Replicate()
    return p


def same_as_input_rule(schema: OpSchema) -> OutputSharding:
    """Output exactly follows input 0 (detach/clone/contiguous/alias)."""
    s = schema.specs[0]
    dtype = s.dtype
    kd = schema.kwargs_schema.get("dtype")
    if isinstance(kd, torch.dtype):
        dtype = kd
    osp = out_spec(s.mesh, s.placements, tuple(s.shape), dtype)
    return OutputSharding(output_spec=osp, input_targets=None)


def replicate_all_rule(schema: OpSchema) -> OutputSharding:
    from ..dispatch import _REPLICATE_OUT

    mesh = schema.mesh
    rep = tuple(Replicate() for _ in range(mesh.ndim))
    return OutputSharding(
        output_spec=_REPLICATE_OUT,
        input_targets=[rep for _ in schema.specs],
    )
