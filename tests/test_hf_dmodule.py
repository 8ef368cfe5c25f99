"""TP-parallelize an UNMODIFIED HuggingFace LlamaForCausalLM through
dmodule sharding plans (vescale_amd/models/hf_llama_plan.py) — the
reference's llama2_4D_finetune drop-in capability
(legacy/examples/llama2_4D_finetune/sharding_plan.py), on transformers>=5.

FORK SAFETY (tests/README): transformers must never be imported in the
pytest parent process — the spawned children import it fresh, and the
spawn uses the 'spawn' start method via backend="gloo" child bodies that
only import inside the body.
"""
import pytest
import torch

from tests.common import spawn


def _t_hf_tp_parity(rank, ws, attn="eager"):
    from transformers import LlamaConfig, LlamaForCausalLM

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import Replicate, init_device_mesh
    from vescale_amd.models.hf_llama_plan import hf_llama_tp_plan

    cfg = LlamaConfig(
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        vocab_size=256,
        attn_implementation=attn,
    )
    torch.manual_seed(7)
    ref = LlamaForCausalLM(cfg)
    ids = torch.randint(0, 256, (2, 8))
    labels = torch.randint(0, 256, (2, 8))
    ref_out = ref(input_ids=ids, labels=labels)
    ref_out.loss.backward()

    torch.manual_seed(7)
    m = LlamaForCausalLM(cfg)
    mesh = init_device_mesh("cpu", (ws,))
    m = parallelize_module(m, mesh, hf_llama_tp_plan())

    out = m(input_ids=ids, labels=labels)
    logits = out.logits
    if hasattr(logits, "redistribute"):
        logits = logits.redistribute(placements=[Replicate()])._local_tensor
    assert torch.allclose(logits, ref_out.logits, atol=2e-5)

    loss = out.loss
    lval = float(loss.full_tensor() if hasattr(loss, "full_tensor") else loss)
    assert abs(lval - float(ref_out.loss)) < 2e-5

    # backward through the TP model: every param gets a grad, and the
    # replicated embedding grad matches the single-process reference
    loss.backward()
    params = dict(m.named_parameters())
    for n, p in params.items():
        assert p.grad is not None, n
    ref_params = dict(ref.named_parameters())
    g = params["model.embed_tokens.weight"].grad
    g = g._local_tensor if hasattr(g, "_local_tensor") else g
    rg = ref_params["model.embed_tokens.weight"].grad
    assert torch.allclose(g, rg, atol=5e-5), (g - rg).abs().max()

    # sharded grad parity: q_proj grad's full tensor matches reference
    from vescale_amd.dtensor import DTensor

    qg = params["model.layers.0.self_attn.q_proj.weight"].grad
    if isinstance(qg, DTensor):
        qg = qg.full_tensor()
    rqg = ref_params["model.layers.0.self_attn.q_proj.weight"].grad
    assert torch.allclose(qg, rqg, atol=5e-5), (qg - rqg).abs().max()


@pytest.mark.parametrize("attn", ["eager", "sdpa"])
def test_hf_llama_tp_parity(attn):
    # sdpa exercises the head-sharded _scaled_dot_product rules end-to-end
    pytest.importorskip("transformers")
    spawn(2, _t_hf_tp_parity, attn)


def _t_hf_tp_sp_parity(rank, ws):
    """sp=True: sequence-sharded (Shard(1)) activations between decoder
    layers — the Megatron-SP boundary — on the unmodified HF model."""
    import torch as _t

    from transformers import LlamaConfig, LlamaForCausalLM

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import Replicate, init_device_mesh
    from vescale_amd.models.hf_llama_plan import hf_llama_tp_plan

    cfg = LlamaConfig(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, vocab_size=256,
        attn_implementation="eager",
    )
    _t.manual_seed(7)
    ref = LlamaForCausalLM(cfg)
    ids = _t.randint(0, 256, (2, 8))
    with _t.no_grad():
        rl = ref(input_ids=ids).logits
    _t.manual_seed(7)
    m = LlamaForCausalLM(cfg)
    mesh = init_device_mesh("cpu", (ws,))
    m = parallelize_module(m, mesh, hf_llama_tp_plan(sp=True))
    with _t.no_grad():
        out = m(input_ids=ids).logits
    if hasattr(out, "redistribute"):
        out = out.redistribute(placements=[Replicate()])._local_tensor
    assert _t.allclose(out, rl, atol=2e-5), (out - rl).abs().max()


def test_hf_llama_tp_sp_parity():
    pytest.importorskip("transformers")
    spawn(2, _t_hf_tp_sp_parity)


def _t_hf_mixtral_tp_parity(rank, ws):
    from transformers import MixtralConfig, MixtralForCausalLM

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import Replicate, init_device_mesh
    from vescale_amd.models.hf_mixtral_plan import hf_mixtral_tp_plan

    cfg = MixtralConfig(
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        vocab_size=256,
        num_local_experts=4,
        num_experts_per_tok=2,
        attn_implementation="eager",
    )
    torch.manual_seed(7)
    ref = MixtralForCausalLM(cfg)
    ids = torch.randint(0, 256, (2, 8))
    with torch.no_grad():
        ref_logits = ref(input_ids=ids).logits

    torch.manual_seed(7)
    m = MixtralForCausalLM(cfg)
    mesh = init_device_mesh("cpu", (ws,))
    m = parallelize_module(m, mesh, hf_mixtral_tp_plan())
    out = m(input_ids=ids).logits
    if hasattr(out, "redistribute"):
        out = out.redistribute(placements=[Replicate()])._local_tensor
    assert torch.allclose(out, ref_logits, atol=2e-5), (out - ref_logits).abs().max()


def test_hf_mixtral_tp_parity():
    pytest.importorskip("transformers")
    spawn(2, _t_hf_mixtral_tp_parity)


def _t_hf_modeloutput_dict_plan(rank, ws):
    """Dict output plans convert fields of a REAL HF ModelOutput return
    (reference test_obj_return.py's dict_like/mixed cases)."""
    import torch.nn as nn
    from transformers.modeling_outputs import BaseModelOutput

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import DTensor, Replicate, Shard, init_device_mesh

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 8)

        def forward(self, x):
            h = self.fc(x)
            return BaseModelOutput(last_hidden_state=h, hidden_states=(h + 1,))

    mesh = init_device_mesh("cpu", (ws,))
    net = parallelize_module(
        Net(), mesh,
        {"parameter": {}, "forward": {"output": {"last_hidden_state": [Shard(0)]}}},
    )
    out = net(torch.randn(2 * ws, 8))
    assert isinstance(out, BaseModelOutput)
    assert isinstance(out.last_hidden_state, DTensor)
    assert out.last_hidden_state.placements[0].is_shard(0)
    # unplanned field untouched by the plan: params default to Replicate
    # DTensors, so it's a DTensor too — but NOT redistributed to Shard
    h0 = out.hidden_states[0]
    assert isinstance(h0, DTensor) and h0.placements[0].is_replicate()


def test_hf_modeloutput_dict_plan():
    pytest.importorskip("transformers")
    spawn(2, _t_hf_modeloutput_dict_plan)
