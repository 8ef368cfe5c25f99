"""SDPA sharding rules: head- or batch-sharded attention passes through
locally (TP head-parallel attention).  Parity: the sdpa flash/efficient
rules noted at legacy/vescale/dtensor/README.md:68-69."""
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


def _qkv_base(schema: OpSchema):
    q = schema.specs[0]
    mesh = q.mesh
    # q/k/v are (B, H, S, D): allow Shard(0) or Shard(1); anything else
    # (S/D-sharded, partial, ragged) replicates.
    base: List[Placement] = []
    for p in q.placements:
        if isinstance(p, Shard) and p.dim in (0, 1):
            base.append(p)
        else:
            base.append(Replicate())
    return q, mesh, tuple(base)


def sdpa_flash_rule(schema: OpSchema) -> OutputSharding:
    q, mesh, base = _qkv_base(schema)
    n = len(schema.specs)
    targets = [base] * n
    B, H, S, D = q.shape
    out0 = out_spec(mesh, base, (B, H, S, D), q.dtype)
    lse = out_spec(mesh, base, (B, H, S), torch.float32)
    outs: List[Optional[DTensorSpec]] = [out0, lse]
    # remaining outputs (cum_seq_q/k, max_q/k, rng state, debug mask) are
    # rank-local bookkeeping: leave unwrapped
    outs += [None] * 7
    return OutputSharding(outs, list(targets))


def sdpa_flash_bwd_rule(schema: OpSchema) -> OutputSharding:
    # grad_out, q, k, v, out, lse, ... -> (dq, dk, dv)
    q = schema.specs[1]
    mesh = q.mesh
    base: List[Placement] = []
    for p in q.placements:
        if isinstance(p, Shard) and p.dim in (0, 1):
            base.append(p)
        else:
            base.append(Replicate())
    base = tuple(base)
    targets = [base] * len(schema.specs)
    k = schema.specs[2]
    v = schema.specs[3]
    dq = out_spec(mesh, base, tuple(q.shape), q.dtype)
    dk = out_spec(mesh, base, tuple(k.shape), k.dtype)
    dv = out_spec(mesh, base, tuple(v.shape), v.dtype)
    return OutputSharding([dq, dk, dv], list(targets))


def sdpa_cpu_rule(schema: OpSchema) -> OutputSharding:
    # _scaled_dot_product_flash_attention_for_cpu -> (out, lse)
    q, mesh, base = _qkv_base(schema)
    n = len(schema.specs)
    B, H, S, D = q.shape
    out0 = out_spec(mesh, base, (B, H, S, D), q.dtype)
    lse = out_spec(mesh, base, (B, H, S), torch.float32)
    return OutputSharding([out0, lse], [base] * n)


def sdpa_cpu_bwd_rule(schema: OpSchema) -> OutputSharding:
    q = schema.specs[1]
    mesh = q.mesh
    base = []
    for p in q.placements:
        if isinstance(p, Shard) and p.dim in (0, 1):
            base.append(p)
        else:
            base.append(Replicate())
    base = tuple(base)
    k, v = schema.specs[2], schema.specs[3]
    dq = out_spec(mesh, base, tuple(q.shape), q.dtype)
    dk = out_spec(mesh, base, tuple(k.shape), k.dtype)
    dv = out_spec(mesh, base, tuple(v.shape), v.dtype)
    return OutputSharding([dq, dk, dv], [base] * len(schema.specs))


def register(dispatcher):
    if hasattr(aten, "_scaled_dot_product_efficient_attention"):
        # same sharding semantics as the flash variant (head/batch sharded)
        dispatcher.register_rule(
            aten._scaled_dot_product_efficient_attention.default, sdpa_flash_rule
        )
        dispatcher.register_rule(
            aten._scaled_dot_product_efficient_attention_backward.default,
            sdpa_flash_bwd_rule,
        )
    if hasattr(aten, "_scaled_dot_product_flash_attention"):
        dispatcher.register_rule(
            aten._scaled_dot_product_flash_attention.default, sdpa_flash_rule
        )
        dispatcher.register_rule(
            aten._scaled_dot_product_flash_attention_backward.default, sdpa_flash_bwd_rule
        )
    if hasattr(aten, "_scaled_dot_product_flash_attention_for_cpu"):
        dispatcher.register_rule(
            aten._scaled_dot_product_flash_attention_for_cpu.default, sdpa_cpu_rule
        )
        dispatcher.register_rule(
            aten._scaled_dot_product_flash_attention_for_cpu_backward.default,
            sdpa_cpu_bwd_rule,
        )
