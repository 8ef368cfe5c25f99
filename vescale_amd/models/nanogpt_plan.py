"""Megatron-style TP/SP sharding plans for nanoGPT.

Parity: legacy/examples/nanogpt_4D_finetune/sharding_plan.py:20-76 —
attention qkv colwise (Shard(0)), out-proj rowwise (Shard(1)); MLP fc
colwise, proj rowwise; block inputs/outputs sequence-sharded (Shard(1))
for SP; LayerNorm runs on the sequence shards.
"""
from vescale_amd.dtensor import InterleavedShard, Replicate, Shard

_R = Replicate()


def nanogpt_tp_plan(sp: bool = True):
    """1-D TP mesh plan.  With sp=True activations between blocks are
    sequence-sharded (Shard(1)); inner attention/MLP activations are
    head-/hidden-sharded by the colwise weights."""
    boundary = [Shard(1)] if sp else [_R]
    param = {
        r"transformer.h.\d+.attn.c_attn.weight": [InterleavedShard(0, 3)],
        r"transformer.h.\d+.attn.c_attn.bias": [InterleavedShard(0, 3)],
        r"transformer.h.\d+.attn.c_proj.weight": [Shard(1)],
        r"transformer.h.\d+.attn.c_proj.bias": [_R],
        r"transformer.h.\d+.mlp.c_fc.weight": [Shard(0)],
        r"transformer.h.\d+.mlp.c_fc.bias": [Shard(0)],
        r"transformer.h.\d+.mlp.c_proj.weight": [Shard(1)],
        r"transformer.h.\d+.mlp.c_proj.bias": [_R],
    }
    forward = {
        r"transformer.wte.input": [[_R]],
        r"transformer.wte.output": [boundary],
        r"transformer.wpe.output": [[_R]],
        r"transformer.h.\d+.input": [boundary],
        r"transformer.h.\d+.attn.input": [[_R]],
        r"transformer.h.\d+.attn.output": [boundary],
        r"transformer.h.\d+.mlp.input": [[_R]],
        r"transformer.h.\d+.mlp.output": [boundary],
        r"transformer.ln_f.output": [[_R]],
    }
    return {"parameter": param, "forward": forward}
