"""Megatron TP/SP sharding plan for an UNMODIFIED HuggingFace Llama.

Parity capability: the reference's llama2_4D_finetune example parallelizes
a stock `transformers` LlamaForCausalLM purely through sharding plans
(legacy/examples/llama2_4D_finetune/sharding_plan.py) — no model edits.
This is the same plan family for transformers >= 5 module names:
attention q/k/v colwise (Shard(0)), o_proj rowwise (Shard(1)); MLP
gate/up colwise, down rowwise; everything else replicated.  Module-IO
redistributions keep decoder-layer boundaries Replicate (sequence
sharding between blocks is available with sp=True, Shard(1) boundaries).

Works with vescale_amd.dmodule.parallelize_module on a 1-D TP mesh; the
model's internal view/transpose/matmul/softmax ops propagate the head
sharding through the dispatcher's rule tables (the attention reshape
takes the non-contiguous reshape fallback in ops/view_ops.py).
"""
from vescale_amd.dtensor import Replicate, Shard

_R = Replicate()


def hf_llama_tp_plan(sp: bool = False):
    boundary = [Shard(1)] if sp else [_R]
    param = {
        r"model.layers.\d+.self_attn.q_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.k_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.v_proj.weight": [Shard(0)],
        r"model.layers.\d+.self_attn.o_proj.weight": [Shard(1)],
        r"model.layers.\d+.mlp.gate_proj.weight": [Shard(0)],
        r"model.layers.\d+.mlp.up_proj.weight": [Shard(0)],
        r"model.layers.\d+.mlp.down_proj.weight": [Shard(1)],
    }
    forward = {
        r"model.embed_tokens.input": [[_R]],
        r"model.layers.\d+.input": [boundary],
        r"model.layers.\d+.self_attn.input": [[_R]],
        r"model.layers.\d+.self_attn.output": [boundary],
        r"model.layers.\d+.mlp.input": [[_R]],
        r"model.layers.\d+.mlp.output": [boundary],
        r"model.norm.output": [[_R]],
        r"lm_head.input": [[_R]],
    }
    return {"parameter": param, "forward": forward}
