"""Embedding sharding (parity: legacy/vescale/dtensor/ops/embedding_ops.py —
vocab-parallel path: range mask + Partial + allreduce-by-placement)."""
from __future__ import annotations

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


def _handler_embedding(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    weight, indices = args[0], args[1]
    rest = args[2:]
    if not isinstance(weight, DTensor):
        return op(*args, **kwargs)
    wspec = weight._spec
    mesh = wspec.mesh
    ispec = indices._spec if isinstance(indices, DTensor) else None
    local_w = weight._local_tensor
    local_i = indices._local_tensor if isinstance(indices, DTensor) else indices

    out_pl = []
    vocab_dims = []
    for md, p in enumerate(wspec.placements):
        ip = ispec.placements[md] if ispec is not None else Replicate()
        if isinstance(p, Shard) and p.dim == 0:
            vocab_dims.append(md)
            out_pl.append(Partial("sum"))
        elif isinstance(p, Shard) and p.dim == 1:
            # hidden-sharded embedding -> out sharded on last dim
            out_pl.append(Shard(local_i.ndim))
        elif isinstance(ip, (Shard, InterleavedShard)):
            out_pl.append(ip)
        else:
            out_pl.append(Replicate())

    if vocab_dims:
        # vocab-parallel: mask out-of-range, clamp, lookup, zero masked rows
        md = vocab_dims[0]
        idx_rank = mesh.get_local_rank(md)
        w = mesh.size(md)
        vocab = wspec.shape[0]
        start = Shard.chunk_offset(vocab, w, idx_rank)
        n = local_w.shape[0]
        mask = (local_i < start) | (local_i >= start + n)
        shifted = (local_i - start).clamp_(0, max(0, n - 1))
        out = op(local_w, shifted, *rest, **kwargs)
        out = out.masked_fill(mask.unsqueeze(-1), 0)
    else:
        out = op(local_w, local_i, *rest, **kwargs)

    shape = tuple((ispec.shape if ispec is not None else local_i.shape)) + (wspec.shape[1],)
    osp = out_spec(mesh, out_pl, shape, out.dtype)
    return DTensor(out, osp, requires_grad=out.requires_grad)


def _handler_embedding_dense_backward(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    grad_output, indices, num_weights = args[0], args[1], args[2]
    rest = args[3:]
    if not isinstance(grad_output, DTensor):
        return op(*args, **kwargs)
    gspec = grad_output._spec
    mesh = gspec.mesh
    local_g = grad_output._local_tensor
    local_i = indices._local_tensor if isinstance(indices, DTensor) else indices

    # batch-sharded grads -> Partial grad weight; hidden-sharded -> Shard(1);
    # Partial grad_output (vocab-parallel fwd) must be handled by masking in
    # the VocabParallelEmbedding module path — generic path reduces first.
    out_pl = []
    need_rep = []
    g_ndim = grad_output._spec.ndim
    coord = mesh.get_coordinate()
    for md, p in enumerate(gspec.placements):
        if isinstance(p, Partial):
            need_rep.append(md)
            out_pl.append(Replicate())
        elif isinstance(p, Shard) and p.dim == g_ndim - 1:
            out_pl.append(Shard(1))
        elif isinstance(p, (Shard, InterleavedShard)):
            out_pl.append(Partial("sum"))
            # indices must be sliced to match the grad's batch/seq shard
            if (
                not isinstance(indices, DTensor)
                and isinstance(p, Shard)
                and p.dim < local_i.ndim
                and local_i.shape[p.dim] == gspec.shape[p.dim]
            ):
                from ..placement_types import Shard as _S

                total = local_i.shape[p.dim]
                w = mesh.size(md)
                off = _S.chunk_offset(total, w, coord[md])
                sz = _S.chunk_size(total, w, coord[md])
                local_i = local_i.narrow(p.dim, off, sz)
        else:
            out_pl.append(Replicate())
    if need_rep:
        grad_output = grad_output.redistribute(
            placements=[
                Replicate() if md in need_rep else p
                for md, p in enumerate(gspec.placements)
            ]
        )
        local_g = grad_output._local_tensor
    out = op(local_g, local_i, num_weights, *rest, **kwargs)
    osp = out_spec(mesh, out_pl, (num_weights, out.shape[1]), out.dtype)
    return DTensor(out, osp, requires_grad=out.requires_grad)


def register(dispatcher):
    dispatcher.register_handler(aten.embedding.default, _handler_embedding)
    dispatcher.register_handler(
        aten.embedding_dense_backward.default, _handler_embedding_dense_backward
    )
