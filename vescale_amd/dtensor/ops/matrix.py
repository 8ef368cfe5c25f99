"""Matmul-family sharding strategies (parity: legacy/vescale/dtensor/ops/
matrix_ops.py, vescale/dtensor/_ops/_matrix_ops.py).

Strategy selection: for each mesh dim, enumerate the valid einsum placements
for mm/bmm and pick the one with the lowest redistribution cost from the
inputs' current placements.  TP partial-sums (S(1) x S(0) -> Partial) are
first-class — that's the row-parallel Linear pattern whose allreduce the
DModule hooks then place.
"""
from __future__ import annotations

from typing import List, Tuple

import torch

from .._collective_utils import allgather_cost, allreduce_cost
from .._dtensor_spec import DTensorSpec
from .._op_schema import OpSchema, OutputSharding
from ..placement_types import InterleavedShard, Partial, Placement, Replicate, Shard
from .common import out_spec

aten = torch.ops.aten


def _is_cands(specs, b_idx: int, out_ndim: int, bias_idx=None):
    """Extra candidates for InterleavedShard weights (packed-QKV colwise
    TP): b placed IS(1, y) -> out IS(last, y) with zero movement."""
    out = []
    b = specs[b_idx]
    for md in range(b.mesh.ndim):
        p = b.placements[md]
        if isinstance(p, InterleavedShard) and p.dim == b.ndim - 1:
            cand_in = []
            for i, s in enumerate(specs):
                if i == b_idx:
                    cand_in.append(p)
                elif bias_idx is not None and i == bias_idx:
                    cand_in.append(InterleavedShard(s.ndim - 1, p.interleaved_size))
                else:
                    cand_in.append(Replicate())
            out.append((tuple(cand_in), InterleavedShard(out_ndim - 1, p.interleaved_size)))
    return out


def _move_cost(cur: Placement, want: Placement, nbytes_gb: float, w: int) -> float:
    if cur == want:
        return 0.0
    if isinstance(cur, Shard) and isinstance(want, Replicate):
        return allgather_cost(nbytes_gb, w)
    if isinstance(cur, Partial):
        return allreduce_cost(nbytes_gb, w)
    if isinstance(cur, Replicate):
        # local slice — cheap, but never FREE: an all-replicate op must not
        # spontaneously shard its output (placement-preserving tie-break)
        return 1e-3
    return allgather_cost(nbytes_gb, w) * 1.5


def _pick(
    mesh, specs: List[DTensorSpec], candidates: List[Tuple[Tuple[Placement, ...], Placement]]
) -> Tuple[List[List[Placement]], List[Placement]]:
    """candidates: per-mesh-dim list of ((input placements...), out placement).
    Greedy per mesh dim: choose candidate minimizing redistribution cost,
    with a penalty for Partial outputs (deferred allreduce has a cost too,
    but is often the right TP answer — penalize lightly)."""
    targets = [list(s.placements) for s in specs]
    outs: List[Placement] = []
    for md in range(mesh.ndim):
        w = mesh.size(md)
        best, best_cost = None, None
        for cand_in, cand_out in candidates:
            cost = 0.0
            for i, s in enumerate(specs):
                cost += _move_cost(s.placements[md], cand_in[i], s.bytes() / 1e9, w)
            if isinstance(cand_out, Partial):
                # output will eventually be reduced; price half an allreduce
                cost += allreduce_cost(specs[0].bytes() / 1e9, w) * 0.25
            if best_cost is None or cost < best_cost:
                best, best_cost = (cand_in, cand_out), cost
        cin, cout = best
        for i in range(len(specs)):
            targets[i][md] = cin[i]
        outs.append(cout)
    return targets, outs


def mm_rule(schema: OpSchema) -> OutputSharding:
    a, b = schema.specs[0], schema.specs[1]
    mesh = a.mesh
    M, K = a.shape
    K2, N = b.shape
    R = Replicate()
    cands = [
        ((Shard(0), R), Shard(0)),
        ((R, Shard(1)), Shard(1)),
        ((Shard(1), Shard(0)), Partial("sum")),
        ((R, R), R),
    ] + _is_cands([a, b], 1, 2)
    targets, outs = _pick(mesh, [a, b], cands)
    osp = out_spec(mesh, outs, (M, N), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def addmm_rule(schema: OpSchema) -> OutputSharding:
    bias, a, b = schema.specs[0], schema.specs[1], schema.specs[2]
    mesh = a.mesh
    M, K = a.shape
    _, N = b.shape
    R = Replicate()
    # no Partial output for addmm (bias would be added w times)
    cands = [
        ((R, Shard(0), R), Shard(0)),
        ((Shard(1) if len(bias.shape) >= 1 else R, R, Shard(1)), Shard(1)),
        ((R, R, R), R),
    ]
    # bias may broadcast (shape (N,) or (M,N)); Shard(1) of out maps to
    # Shard(len(bias.shape)-1) of bias
    def fix_bias(p: Placement) -> Placement:
        if isinstance(p, Shard):
            return Shard(len(bias.shape) - 1)
        return p

    cands = [((fix_bias(ci[0]), ci[1], ci[2]), co) for ci, co in cands]
    cands += _is_cands([bias, a, b], 2, 2, bias_idx=0)
    targets, outs = _pick(mesh, [bias, a, b], cands)
    osp = out_spec(mesh, outs, (M, N), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def bmm_rule(schema: OpSchema) -> OutputSharding:
    a, b = schema.specs[0], schema.specs[1]
    mesh = a.mesh
    B, M, K = a.shape
    _, _, N = b.shape
    R = Replicate()
    cands = [
        ((Shard(0), Shard(0)), Shard(0)),
        ((Shard(1), R), Shard(1)),
        ((R, Shard(2)), Shard(2)),
        ((Shard(2), Shard(1)), Partial("sum")),
        ((R, R), R),
    ]
    targets, outs = _pick(mesh, [a, b], cands)
    osp = out_spec(mesh, outs, (B, M, N), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def baddbmm_rule(schema: OpSchema) -> OutputSharding:
    bias, a, b = schema.specs
    mesh = a.mesh
    B, M, K = a.shape
    _, _, N = b.shape
    R = Replicate()
    cands = [
        ((Shard(0), Shard(0), Shard(0)), Shard(0)),
        ((R, R, R), R),
    ]
    targets, outs = _pick(mesh, [bias, a, b], cands)
    osp = out_spec(mesh, outs, (B, M, N), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def mv_rule(schema: OpSchema) -> OutputSharding:
    a, v = schema.specs
    mesh = a.mesh
    M, K = a.shape
    R = Replicate()
    cands = [
        ((Shard(0), R), Shard(0)),
        ((Shard(1), Shard(0)), Partial("sum")),
        ((R, R), R),
    ]
    targets, outs = _pick(mesh, [a, v], cands)
    osp = out_spec(mesh, outs, (M,), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def dot_rule(schema: OpSchema) -> OutputSharding:
    a, b = schema.specs
    mesh = a.mesh
    cands = [
        ((Shard(0), Shard(0)), Partial("sum")),
        ((Replicate(), Replicate()), Replicate()),
    ]
    targets, outs = _pick(mesh, [a, b], cands)
    osp = out_spec(mesh, outs, (), a.dtype)
    return OutputSharding(osp, [tuple(t) for t in targets])


def register(dispatcher):
    dispatcher.register_rule(aten.mm.default, mm_rule)
    dispatcher.register_rule(aten.addmm.default, addmm_rule)
    dispatcher.register_rule(aten.bmm.default, bmm_rule)
    dispatcher.register_rule(aten.baddbmm.default, baddbmm_rule)
    dispatcher.register_rule(aten.mv.default, mv_rule)
    dispatcher.register_rule(aten.dot.default, dot_rule)
