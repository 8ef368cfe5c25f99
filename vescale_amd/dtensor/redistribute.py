"""Placement-to-placement conversion — *the* communication generator.

Every (src placement -> dst placement) transition on a mesh dim lowers to one
RCCL collective over xGMI:

    S(d) -> R      : all-gather            (uneven-aware, pad-free)
    P    -> R      : all-reduce
    P    -> S(d)   : reduce-scatter        (pad if not divisible)
    S(a) -> S(b)   : all-to-all
    R    -> S(d)   : local slice           (no comm)
    R    -> P      : keep-on-rank0/zero    (no comm)
    RS   -> R      : uneven all-gather + reshape  (the FSDP unshard path)
    RS   -> RS'    : interval-intersection all-to-all
    R    -> RS     : local flat narrow     (no comm)
    IS(d)-> R      : all-gather + de-interleave

Parity: legacy/vescale/dtensor/redistribute.py:51-658 and the ragged paths at
vescale/dtensor/_redistribute.py:48-120.  Design is transition-table-driven
rather than a port.
"""
from __future__ import annotations

from typing import List, Sequence

import torch

from . import _collective_utils as cc
from ._dtensor_spec import DTensorSpec
from .device_mesh import DeviceMesh
from .placement_types import (
    _StridedRaggedShard,
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
)


def _shard_sizes_on_dim(total: int, num_chunks: int) -> List[int]:
    return [Shard.chunk_size(total, num_chunks, i) for i in range(num_chunks)]


def _logical_shape_at(spec: DTensorSpec, upto_placements: Sequence[Placement], coord) -> List[int]:
    """Global shape reduced by shard placements OTHER than the one being
    transitioned — gives per-mesh-dim size accounting on nested sharding."""
    shape = list(spec.shape)
    for mesh_dim, p in enumerate(upto_placements):
        if isinstance(p, InterleavedShard):
            shape = list(p.local_shape(shape, spec.mesh.size(mesh_dim), coord[mesh_dim]))
        elif isinstance(p, Shard):
            shape = list(p.local_shape(shape, spec.mesh.size(mesh_dim), coord[mesh_dim]))
    return shape


def redistribute_local_tensor(
    local: torch.Tensor,
    current_spec: DTensorSpec,
    target_spec: DTensorSpec,
    *,
    async_op: bool = False,
) -> torch.Tensor:
    """Convert the local tensor from current_spec's layout to target_spec's.
    Both specs must be on the same mesh."""
    assert current_spec.mesh == target_spec.mesh, "cross-mesh redistribute: use CrossMeshRedistribute"
    mesh = current_spec.mesh
    coord = mesh.get_coordinate()
    if coord is None:
        return local

    cur: List[Placement] = list(current_spec.placements)
    tgt: List[Placement] = list(target_spec.placements)
    if cur == tgt:
        return local

    # Safety-aware planner for nested sharding (placements apply outermost
    # mesh dim first).  A transition on mesh dim md touching tensor dim d is
    # UNSAFE while an inner mesh dim (> md) still shards d — gathering or
    # slicing there would interleave chunk-of-chunk layouts.  When no safe
    # transition exists, unshard the innermost conflicting dim first.
    all_dims = set(range(len(current_spec.shape)))

    def pdims(p: Placement) -> set:
        """Tensor dims a placement's layer entangles.  A strided-ragged
        local is a flat view of the inner chunk, so it entangles ALL dims."""
        if isinstance(p, _StridedRaggedShard):
            return set(all_dims)
        if isinstance(p, RaggedShard):
            return set(p.dims)
        if isinstance(p, (Shard, InterleavedShard)):
            return {p.dim}
        return set()

    def lkey(md: int, p: Placement):
        # physical application order of a layer: mesh-dim order, except
        # _StridedRaggedShard which composes AFTER every other mesh dim
        # (it flattens the inner chunk) — see api.distribute_tensor
        return (1 if isinstance(p, _StridedRaggedShard) else 0, md)

    def is_safe(md: int) -> bool:
        """A transition peels cur[md] then applies tgt[md].  Peeling is
        outermost-first: no overlapping existing layer may sit above
        cur[md].  Applying goes above existing layers only (never tuck a
        new layer under an outer one), and an overlapping pending target
        layer that is MORE inner must be applied first."""
        pd = pdims(cur[md])
        for k in range(mesh.ndim):
            if k == md:
                continue
            if pd and pdims(cur[k]) & pd and lkey(k, cur[k]) > lkey(md, cur[md]):
                return False
        ad = pdims(tgt[md])
        if ad:
            for k in range(mesh.ndim):
                if k == md:
                    continue
                if pdims(cur[k]) & ad and lkey(k, cur[k]) > lkey(md, tgt[md]):
                    return False
                if (
                    cur[k] != tgt[k]
                    and pdims(tgt[k]) & ad
                    and lkey(k, tgt[k]) < lkey(md, tgt[md])
                ):
                    return False
                if (
                    cur[k] != tgt[k]
                    and pdims(cur[k]) & ad
                    and lkey(k, cur[k]) < lkey(md, tgt[md])
                ):
                    # my new layer would sit INNER to an existing outer
                    # layer that still has to be peeled — peeling it later
                    # would be blocked by me (planner-loop guard, caught by
                    # the property soak: [S(0),R] -> [SRS, S(0)])
                    return False
        return True

    new_local = local
    guard = 0
    while cur != tgt:
        guard += 1
        assert guard <= 4 * mesh.ndim + 4, f"redistribute planner stuck: {cur} -> {tgt}"
        progressed = False
        for md in range(mesh.ndim):
            if cur[md] == tgt[md]:
                continue
            if is_safe(md):
                new_local = _transition(
                    new_local, current_spec, cur, md, cur[md], tgt[md], coord,
                    async_op=async_op,
                )
                cur[md] = tgt[md]
                progressed = True
        if not progressed:
            # unshard the innermost dim that blocks some pending transition
            blocked = [md for md in range(mesh.ndim) if cur[md] != tgt[md]]
            victim = None
            for md in blocked:
                need = pdims(cur[md]) | pdims(tgt[md])
                # candidates INCLUDE md itself: when md's own layer is the
                # outermost (e.g. SRS) the right first move is to peel IT,
                # not an inner layer beneath it (peeling under a live
                # strided-ragged flat view scrambles values — caught by the
                # planner property soak)
                cands = [k for k in range(mesh.ndim) if pdims(cur[k]) & need]
                if cands:
                    victim = max(cands, key=lambda k: lkey(k, cur[k]))
                    break
            assert victim is not None, f"redistribute deadlock: {cur} -> {tgt}"
            new_local = _transition(
                new_local, current_spec, cur, victim, cur[victim], Replicate(), coord,
            )
            cur[victim] = Replicate()
    return new_local


def _transition(
    local: torch.Tensor,
    spec: DTensorSpec,
    cur_placements: List[Placement],
    mesh_dim: int,
    c: Placement,
    t: Placement,
    coord,
    *,
    async_op: bool = False,
) -> torch.Tensor:
    mesh = spec.mesh
    w = mesh.size(mesh_dim)
    my = coord[mesh_dim]
    others = [p if i != mesh_dim else Replicate() for i, p in enumerate(cur_placements)]

    # ---------------- source Replicate ----------------
    if isinstance(c, Replicate):
        if isinstance(t, InterleavedShard):
            return t.split_tensor(local, w)[my]
        if isinstance(t, RaggedShard):
            return t.split_tensor(local, w)[my]
        if isinstance(t, Shard):
            return c_split_replicate(local, t, w, my)
        if isinstance(t, Partial):
            return local if my == 0 else torch.zeros_like(local)

    # ---------------- source Partial ----------------
    if isinstance(c, Partial):
        if isinstance(t, Replicate):
            out = local.clone(memory_format=torch.contiguous_format)
            work = cc.mesh_all_reduce(out, mesh, c.reduce_op, mesh_dim, async_op=async_op)
            if async_op and work is not None:
                work.wait()
            return out
        if isinstance(t, Shard):
            # pad to divisible on shard dim, reduce_scatter, unpad
            d = t.dim
            total = local.size(d)
            sizes = _shard_sizes_on_dim(total, w)
            pad_total = max(sizes[0] * w, total) if sizes else 0
            x = local
            if pad_total != total:
                pad_shape = list(local.shape)
                pad_shape[d] = pad_total - total
                x = torch.cat([local, local.new_zeros(pad_shape)], dim=d)
            out = cc.mesh_reduce_scatter(x, mesh, c.reduce_op, mesh_dim, scatter_dim=d)
            if out.size(d) != sizes[my]:
                out = out.narrow(d, 0, sizes[my]).contiguous()
            return out
        if isinstance(t, RaggedShard):
            # P -> RS: reduce_scatter over the ragged unit ranges.
            # Lower as allreduce + local narrow for correctness; the FSDP
            # engine uses the fused grad reduce-scatter path instead.
            out = local.contiguous()
            cc.mesh_all_reduce(out, mesh, c.reduce_op, mesh_dim)
            return t.split_tensor(out, w)[my]
        if isinstance(t, Partial):
            return local  # reduce-op change unsupported; treat as same

    # ---------------- source Shard ----------------
    if isinstance(c, InterleavedShard):
        if isinstance(t, Replicate) or (isinstance(t, Shard) and not isinstance(t, InterleavedShard)):
            # gather the interleaved chunks then stitch: chunks are
            # [..., IS, inner/w, ...] pieces
            d = c.dim
            logical = _logical_shape_at(spec, others, coord)
            full_inner = logical[d] // c.interleaved_size
            t_loc = local.reshape(
                local.shape[:d] + (c.interleaved_size, full_inner // w) + local.shape[d + 1 :]
            )
            gathered = cc.mesh_all_gather(t_loc, mesh, mesh_dim, gather_dim=d + 1)
            out = gathered.reshape(local.shape[:d] + (logical[d],) + local.shape[d + 1 :])
            if isinstance(t, Replicate):
                return out
            return c_split_replicate(out, t, w, coord[mesh_dim])
        # any other target: go through Replicate
        rep = _transition(local, spec, cur_placements, mesh_dim, c, Replicate(), coord)
        return _transition(rep, spec, cur_placements, mesh_dim, Replicate(), t, coord)

    if isinstance(c, RaggedShard):
        if isinstance(t, Replicate):
            logical = _logical_shape_at(spec, others, coord)
            un = c.unit_numel(logical)
            sizes = [u * un for u in c.local_units]
            flat = cc.mesh_all_gather(local.reshape(-1), mesh, mesh_dim, gather_dim=0, sizes=sizes)
            return RaggedShard.reconstruct(flat, logical)
        if isinstance(t, RaggedShard):
            return _ragged_to_ragged(local, c, t, spec, others, coord, mesh_dim)
        # RS -> S / P: go through Replicate
        rep = _transition(local, spec, cur_placements, mesh_dim, c, Replicate(), coord)
        return _transition(rep, spec, cur_placements, mesh_dim, Replicate(), t, coord)

    if isinstance(c, Shard):
        d = c.dim
        logical = _logical_shape_at(spec, others, coord)
        sizes = _shard_sizes_on_dim(logical[d], w)
        if local.size(d) != sizes[my]:
            # custom uneven layout (from_local support_uneven): agree on the
            # TRUE per-rank sizes (reference gather_local_tensor_shape)
            import torch.distributed as dist

            shapes: list = [None] * w
            dist.all_gather_object(shapes, int(local.size(d)), group=mesh.get_group(mesh_dim))
            sizes = [int(s) for s in shapes]
        if isinstance(t, Replicate):
            even = all(s == sizes[0] for s in sizes)
            return cc.mesh_all_gather(
                local, mesh, mesh_dim, gather_dim=d, sizes=None if even else sizes
            )
        if isinstance(t, Shard) and not isinstance(t, InterleavedShard) and t.dim != d:
            return _shard_to_shard_alltoall(local, c, t, sizes, mesh, mesh_dim, my, logical)
        if isinstance(t, Partial):
            raise NotImplementedError("Shard -> Partial is not a meaningful transition")
        if isinstance(t, (InterleavedShard, RaggedShard)):
            rep = _transition(local, spec, cur_placements, mesh_dim, c, Replicate(), coord)
            return _transition(rep, spec, cur_placements, mesh_dim, Replicate(), t, coord)

    raise NotImplementedError(f"transition {c} -> {t}")


def c_split_replicate(local: torch.Tensor, t: Shard, w: int, my: int) -> torch.Tensor:
    total = local.size(t.dim)
    off = Shard.chunk_offset(total, w, my)
    sz = Shard.chunk_size(total, w, my)
    return local.narrow(t.dim, off, sz).contiguous()


def _shard_to_shard_alltoall(
    local: torch.Tensor,
    c: Shard,
    t: Shard,
    src_sizes: List[int],
    mesh: DeviceMesh,
    mesh_dim: int,
    my: int,
    logical: List[int],
) -> torch.Tensor:
    """S(a) -> S(b) via all_to_all: send chunk j of my dim-b extent to rank j,
    receive others' and cat along dim a."""
    w = mesh.size(mesh_dim)
    dst_sizes = _shard_sizes_on_dim(logical[t.dim], w)
    send = []
    for j in range(w):
        off = Shard.chunk_offset(logical[t.dim], w, j)
        send.append(local.narrow(t.dim, off, dst_sizes[j]).contiguous())
    recv = []
    for j in range(w):
        shape = list(local.shape)
        shape[c.dim] = src_sizes[j]
        shape[t.dim] = dst_sizes[my]
        recv.append(local.new_empty(shape))
    cc.mesh_all_to_all(recv, send, mesh, mesh_dim)
    return torch.cat(recv, dim=c.dim)


def _ragged_to_ragged(
    local: torch.Tensor,
    c: RaggedShard,
    t: RaggedShard,
    spec: DTensorSpec,
    others,
    coord,
    mesh_dim: int,
) -> torch.Tensor:
    """Interval-intersection all_to_all (reference
    vescale/dtensor/placement_types.py:152-211): each rank's flat source
    range [s0,s1) intersects each destination range; per-pair lengths feed
    one uneven all_to_all_single."""
    mesh = spec.mesh
    w = mesh.size(mesh_dim)
    my = coord[mesh_dim]
    logical = _logical_shape_at(spec, others, coord)
    numel = 1
    for s in logical:
        numel *= s

    def ranges(p: RaggedShard):
        un = p.unit_numel(logical)
        out, off = [], 0
        for u in p.local_units:
            out.append((off, off + u * un))
            off += u * un
        return out

    src_r, dst_r = ranges(c), ranges(t)
    s0, s1 = src_r[my]
    in_splits = []
    for j in range(w):
        d0, d1 = dst_r[j]
        in_splits.append(max(0, min(s1, d1) - max(s0, d0)))
    d0, d1 = dst_r[my]
    out_splits = []
    for j in range(w):
        a0, a1 = src_r[j]
        out_splits.append(max(0, min(a1, d1) - max(a0, d0)))
    # source data is already flat-ordered; intersections are contiguous and
    # ordered by j, so a single narrow-concat == the local buffer itself
    out = local.new_empty(sum(out_splits))
    cc.mesh_all_to_all_single(
        out, local.contiguous().view(-1), mesh, mesh_dim,
        output_split_sizes=out_splits, input_split_sizes=in_splits,
    )
    return out


# ---------------------------------------------------------------------------
# autograd wrapper
# ---------------------------------------------------------------------------
class Redistribute(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dtensor, target_placements, async_op=False):
        from .dtensor import DTensor

        prev_spec = dtensor._spec
        mesh = prev_spec.mesh
        target_spec = DTensorSpec(mesh, tuple(target_placements), prev_spec.tensor_meta)
        ctx.prev_spec = prev_spec
        ctx.target_spec = target_spec
        local = redistribute_local_tensor(
            dtensor._local_tensor, prev_spec, target_spec, async_op=async_op
        )
        return DTensor._from_local_spec(local, target_spec, dtensor.requires_grad)

    @staticmethod
    def backward(ctx, grad_output):
        from .dtensor import DTensor

        prev_spec = ctx.prev_spec
        # gradient flows back: Partial targets in forward mean the grad is
        # Replicate; Partial in prev means grad should stay as-is (reference
        # _redistribute.py:174-181 normalizes Partial -> Replicate).
        normalized = tuple(
            Replicate() if isinstance(p, Partial) else p for p in prev_spec.placements
        )
        target = DTensorSpec(prev_spec.mesh, normalized, prev_spec.tensor_meta)
        local = redistribute_local_tensor(
            grad_output._local_tensor, grad_output._spec, target
        )
        out = DTensor._from_local_spec(local, target, grad_output.requires_grad)
        return out, None, None
