"""Tensor-parallel wiring for the Llama family (Megatron f/g operators on
plain tensors — the production TP path the flagship bench composes with
FSDP for the BASELINE 2D config "Llama-3 8B TP x FSDP").

Block weights are TP-sharded (wqkv/w13 by output rows = heads/ffn, wo/w2 by
input cols); embeddings, norms and the LM head stay replicated (all TP
ranks see the same tokens, so their grads agree without extra syncs).
Two collectives per sublayer: f (fwd identity / bwd all-reduce) at the
sublayer input, g (fwd all-reduce / bwd identity) at its output — each a
single RCCL all-reduce of the activation over xGMI.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.distributed as dist


@dataclass
class TPContext:
    group: Optional[dist.ProcessGroup]
    rank: int
    world: int

    @classmethod
    def single(cls):
        return cls(None, 0, 1)

    @classmethod
    def from_group(cls, group):
        if group is None or not dist.is_initialized():
            return cls.single()
        return cls(group, dist.get_rank(group), dist.get_world_size(group))


class _CopyToTP(torch.autograd.Function):
    """Megatron f: forward identity, backward all-reduce over TP."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, g):
        if ctx.group is not None:
            g = g.contiguous()
            dist.all_reduce(g, group=ctx.group)
        return g, None


class _ReduceFromTP(torch.autograd.Function):
    """Megatron g: forward all-reduce over TP, backward identity."""

    @staticmethod
    def forward(ctx, x, group):
        if group is not None:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, g):
        return g, None


def copy_to_tp(x, tp: TPContext):
    if tp.world == 1:
        return x
    return _CopyToTP.apply(x, tp.group)


def reduce_from_tp(x, tp: TPContext):
    if tp.world == 1:
        return x
    return _ReduceFromTP.apply(x, tp.group)


def shard_llama_state_dict(
    full_sd: Dict[str, torch.Tensor], cfg, tp_rank: int, tp_world: int
) -> Dict[str, torch.Tensor]:
    """Slice a TP=1 Llama state dict into rank `tp_rank`'s TP shard
    (test/checkpoint-interop helper).  wqkv rows are [q; k; v] head-blocks;
    w13 rows are [gate; up]."""
    if tp_world == 1:
        return dict(full_sd)
    out = {}
    hd = cfg.head_dim
    hq, hkv = cfg.n_heads, cfg.n_kv_heads
    hq_l, hkv_l = hq // tp_world, hkv // tp_world
    f, f_l = cfg.ffn_dim, cfg.ffn_dim // tp_world
    for k, v in full_sd.items():
        if k.endswith("attn.wqkv.weight"):
            qs = v[: hq * hd]
            ks = v[hq * hd : (hq + hkv) * hd]
            vs = v[(hq + hkv) * hd :]
            out[k] = torch.cat(
                [
                    qs[tp_rank * hq_l * hd : (tp_rank + 1) * hq_l * hd],
                    ks[tp_rank * hkv_l * hd : (tp_rank + 1) * hkv_l * hd],
                    vs[tp_rank * hkv_l * hd : (tp_rank + 1) * hkv_l * hd],
                ]
            ).clone()
        elif k.endswith("attn.wo.weight"):
            out[k] = v[:, tp_rank * hq_l * hd : (tp_rank + 1) * hq_l * hd].clone()
        elif k.endswith("ffn.w13.weight"):
            gates = v[:f]
            ups = v[f:]
            out[k] = torch.cat(
                [
                    gates[tp_rank * f_l : (tp_rank + 1) * f_l],
                    ups[tp_rank * f_l : (tp_rank + 1) * f_l],
                ]
            ).clone()
        elif k.endswith("ffn.w2.weight"):
            out[k] = v[:, tp_rank * f_l : (tp_rank + 1) * f_l].clone()
        else:
            out[k] = v.clone()
    return out
