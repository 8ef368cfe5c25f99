"""Auto-plan policy providers (Megatron-style).

Parity: legacy/vescale/dmp/policies/megatron.py:33-210 + registry.py:22 —
per-module-class plan fragments: MLP alternating colwise/rowwise Linear
pairs, attention qkv Shard(0) (InterleavedShard for packed qkv) + out
Shard(1), LayerNorm SP pass-through, embedding, lm head.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Tuple

import torch.nn as nn

from ..dtensor import InterleavedShard, Replicate, Shard

REGISTRY: Dict[str, Callable] = {}


def register_policy(class_name: str):
    def deco(fn):
        REGISTRY[class_name] = fn
        return fn

    return deco


def _linears_of(mod: nn.Module) -> List[Tuple[str, nn.Linear]]:
    return [(n, m) for n, m in mod.named_modules() if isinstance(m, nn.Linear)]


@register_policy("MLP")
@register_policy("FeedForward")
def mlp_plan_provider(fqn: str, mod: nn.Module, sp: bool):
    """Alternating colwise/rowwise over the module's Linear sequence
    (reference megatron.py:33-58)."""
    param, fwd = {}, {}
    linears = _linears_of(mod)
    boundary = [Shard(1)] if sp else [Replicate()]
    for i, (name, lin) in enumerate(linears):
        key = f"{fqn}.{name}" if name else fqn
        if i % 2 == 0:  # colwise
            param[rf"{key}\.weight"] = [Shard(0)]
            if lin.bias is not None:
                param[rf"{key}\.bias"] = [Shard(0)]
        else:  # rowwise
            param[rf"{key}\.weight"] = [Shard(1)]
            if lin.bias is not None:
                param[rf"{key}\.bias"] = [Replicate()]
    fwd[rf"{fqn}\.input"] = [[Replicate()]]
    fwd[rf"{fqn}\.output"] = [boundary]
    return param, fwd


@register_policy("Attention")
@register_policy("CausalSelfAttention")
def attention_plan_provider(fqn: str, mod: nn.Module, sp: bool):
    """qkv colwise (InterleavedShard(0,3) when packed in one Linear),
    out-proj rowwise (reference megatron.py:90-160)."""
    param, fwd = {}, {}
    boundary = [Shard(1)] if sp else [Replicate()]
    linears = _linears_of(mod)
    out_names = ("c_proj", "wo", "out_proj", "o_proj", "dense")
    for name, lin in linears:
        key = f"{fqn}.{name}"
        if any(name.endswith(o) or name == o for o in out_names):
            param[rf"{key}\.weight"] = [Shard(1)]
            if lin.bias is not None:
                param[rf"{key}\.bias"] = [Replicate()]
        else:
            packed = lin.out_features % (3 * max(1, lin.in_features)) == 0 and lin.out_features == 3 * lin.in_features
            pl = [InterleavedShard(0, 3)] if packed else [Shard(0)]
            param[rf"{key}\.weight"] = pl
            if lin.bias is not None:
                param[rf"{key}\.bias"] = pl
    fwd[rf"{fqn}\.input"] = [[Replicate()]]
    fwd[rf"{fqn}\.output"] = [boundary]
    return param, fwd


@register_policy("LayerNorm")
@register_policy("RMSNorm")
def layernorm_plan_provider(fqn: str, mod: nn.Module, sp: bool):
    """SP: norms run on sequence shards; weights replicated (their Partial
    grads are synced by DModule grad sync).  (reference megatron.py:162)"""
    return {}, {}


@register_policy("Embedding")
def embedding_plan_provider(fqn: str, mod: nn.Module, sp: bool):
    boundary = [Shard(1)] if sp else [Replicate()]
    fwd = {rf"{fqn}\.output": [boundary]} if sp else {}
    return {}, fwd
