"""Megatron TP sharding plan for an UNMODIFIED HuggingFace Mixtral.

Parity capability: legacy/examples/mixtral_4D_benchmark/sharding_plan.py —
a stock `transformers` MixtralForCausalLM parallelized purely through
plans.  Attention like Llama (q/k/v colwise, o rowwise); every expert's
w1/w3 colwise and w2 rowwise, so each expert's FFN is TP-sharded while
the routing (softmax + topk + one_hot + index ops) stays replicated and
propagates through the dispatcher's rule tables — the index/one_hot
handlers in dtensor/ops/extra_ops.py carry the gather/scatter steps.
"""
from vescale_amd.dtensor import Replicate, Shard

_R = Replicate()


def hf_mixtral_tp_plan():
    param = {
        r"model.layers.\d+.self_attn.q_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.k_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.v_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.o_proj.weight": [Shard(1)],
        r"model.layers.\d+.block_sparse_moe.experts.\d+.w1.weight": [Shard(0)],
        r"model.layers.\d+.block_sparse_moe.experts.\d+.w3.weight": [Shard(0)],
        r"model.layers.\d+.block_sparse_moe.experts.\d+.w2.weight": [Shard(1)],
    }
    forward = {
        r"model.embed_tokens.input": [[_R]],
        r"model.layers.\d+.input": [[_R]],
        r"model.layers.\d+.self_attn.input": [[_R]],
        r"model.layers.\d+.self_attn.output": [[_R]],
        r"model.layers.\d+.block_sparse_moe.input": [[_R]],
        r"model.layers.\d+.block_sparse_moe.output": [[_R]],
        r"model.norm.output": [[_R]],
        r"lm_head.input": [[_R]],
    }
    return {"parameter": param, "forward": forward}
