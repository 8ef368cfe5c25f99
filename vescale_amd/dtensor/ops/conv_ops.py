"""Convolution dispatch: batch-parallel conv2d/3d on DTensors.

Parity: legacy/vescale/dtensor/ops/conv_ops.py:21-108 (convolution_rules /
convolution_backward_rules): input batch-sharded Shard(0) with replicated
weight/bias runs locally per rank; grad_input follows the input's batch
sharding, grad_weight/grad_bias come back Partial("sum") over the
batch-sharded mesh dims.  Any other input layout is redistributed to
Replicate first (correctness fallback).
"""
from __future__ import annotations

import torch

from ..placement_types import InterleavedShard, Partial, RaggedShard, Replicate, Shard
from .common import out_spec

aten = torch.ops.aten


def _batch_dims(spec):
    return [
        md for md, p in enumerate(spec.placements)
        if isinstance(p, (Shard, InterleavedShard)) and p.dim == 0
    ]


def _is_bad(spec):
    return any(
        (isinstance(p, (Shard, InterleavedShard)) and p.dim != 0)
        or isinstance(p, (RaggedShard, Partial))
        for p in spec.placements
    )


def _replicate_all(t):
    from ..dtensor import DTensor

    if isinstance(t, DTensor):
        return t.redistribute(placements=[Replicate()] * t._spec.mesh.ndim)
    return t


def _handler_convolution(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    inp, weight, bias = args[0], args[1], args[2]
    if not isinstance(inp, DTensor):
        return op(*args, **kwargs)
    mesh = inp._spec.mesh
    if _is_bad(inp._spec):
        inp = _replicate_all(inp)
    weight = _replicate_all(weight)
    bias = _replicate_all(bias)
    bdims = _batch_dims(inp._spec)
    local = op(
        inp._local_tensor,
        weight._local_tensor if isinstance(weight, DTensor) else weight,
        bias._local_tensor if isinstance(bias, DTensor) else bias,
        *args[3:],
        **kwargs,
    )
    # output: batch dim shards exactly like the input's
    shard_factor = 1
    for md in bdims:
        shard_factor *= mesh.size(md)
    gshape = (local.shape[0] * shard_factor,) + tuple(local.shape[1:])
    pl = [Shard(0) if md in bdims else Replicate() for md in range(mesh.ndim)]
    return DTensor(local, out_spec(mesh, pl, gshape, local.dtype),
                   requires_grad=local.requires_grad)


def _handler_convolution_backward(dispatcher, op, args, kwargs):
    from ..dtensor import DTensor

    grad_out, inp, weight = args[0], args[1], args[2]
    if not isinstance(grad_out, DTensor) and not isinstance(inp, DTensor):
        return op(*args, **kwargs)
    mesh = (grad_out if isinstance(grad_out, DTensor) else inp)._spec.mesh
    if isinstance(inp, DTensor) and _is_bad(inp._spec):
        inp = _replicate_all(inp)
    weight = _replicate_all(weight)
    # batch sharding follows the INPUT; reconcile grad_out to it (autograd
    # may hand us a Replicate stride-0 expand from a scalar-loss backward)
    bdims = _batch_dims(inp._spec) if isinstance(inp, DTensor) else []
    if isinstance(grad_out, DTensor):
        want = [Shard(0) if md in bdims else Replicate()
                for md in range(mesh.ndim)]
        if list(grad_out._spec.placements) != want:
            grad_out = grad_out.redistribute(placements=want)

    def loc(t):
        # oneDNN rejects expanded-stride grads (sum().backward gives a
        # broadcast-expanded grad_out) — make locals contiguous
        lt = t._local_tensor if isinstance(t, DTensor) else t
        return lt.contiguous() if torch.is_tensor(lt) else lt

    gi, gw, gb = op(loc(grad_out), loc(inp), loc(weight), *args[3:], **kwargs)
    outs = []
    # grad_input: batch-sharded like grad_out
    if gi is not None:
        shard_factor = 1
        for md in bdims:
            shard_factor *= mesh.size(md)
        gshape = (gi.shape[0] * shard_factor,) + tuple(gi.shape[1:])
        pl = [Shard(0) if md in bdims else Replicate() for md in range(mesh.ndim)]
        outs.append(DTensor(gi, out_spec(mesh, pl, gshape, gi.dtype),
                            requires_grad=False))
    else:
        outs.append(None)
    # grad_weight / grad_bias: Partial over the batch-sharded dims
    for g in (gw, gb):
        if g is None:
            outs.append(None)
            continue
        pl = [Partial("sum") if md in bdims else Replicate()
              for md in range(mesh.ndim)]
        outs.append(DTensor(g, out_spec(mesh, pl, tuple(g.shape), g.dtype),
                            requires_grad=False))
    return tuple(outs)


def register(dispatcher):
    dispatcher.register_handler(aten.convolution.default, _handler_convolution)
    dispatcher.register_handler(
        aten.convolution_backward.default, _handler_convolution_backward
    )
