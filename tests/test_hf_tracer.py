"""HF-aware pipeline tracer + PARAMETERS split
(vescale_amd/pipe/pipe_parser.py: hf_symbolic_trace / parse_huggingface_model).

Reference capability: legacy/vescale/pipe/tracer.py:93,626 (hf_symbolic_trace
over HFTracer) + pipe_parser.py:146 (PipelineSplitMethodType.PARAMETERS).
transformers >= 5 removed utils.fx, so ours is a native tracer; these tests
run it against a REAL `transformers` LlamaForCausalLM (random init, no
network) — kwargs forward with a **kwargs catch-all, ModelOutput returns,
data-dependent mask helpers.
"""
import pytest
import torch

transformers = pytest.importorskip("transformers")


@pytest.fixture(scope="module")
def hf_llama():
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=4,
        num_attention_heads=4,
        num_key_value_heads=2,
        vocab_size=256,
    )
    torch.manual_seed(0)
    return LlamaForCausalLM(cfg)


def _run_chain(stages, ids):
    cur = (ids,)
    for s in stages:
        cur = s(*cur)
        if not isinstance(cur, (tuple, list)):
            cur = (cur,)
    return cur[0]


def test_hf_trace_parity(hf_llama):
    from vescale_amd.pipe import hf_symbolic_trace

    gm = hf_symbolic_trace(hf_llama)
    ids = torch.randint(0, 256, (2, 8))
    ref = hf_llama(input_ids=ids).logits
    out = gm(ids)
    out = out[0] if isinstance(out, (tuple, list)) else out
    assert torch.allclose(out, ref, atol=1e-6)


@pytest.mark.parametrize("num_stages", [2, 3, 4])
def test_hf_parameters_split_parity(hf_llama, num_stages):
    from vescale_amd.pipe import parse_huggingface_model

    stages = parse_huggingface_model(hf_llama, num_stages)
    assert len(stages) == num_stages
    ids = torch.randint(0, 256, (2, 8))
    ref = hf_llama(input_ids=ids).logits
    out = _run_chain(stages, ids)
    assert torch.allclose(out, ref, atol=1e-6)
    # PARAMETERS criterion: no stage ends up empty, balance within 2.5x
    sizes = [sum(p.numel() for p in s.parameters()) for s in stages]
    assert all(sz > 0 for sz in sizes)
    assert max(sizes) / min(sizes) < 2.5, sizes


def test_hf_split_backward_flows(hf_llama):
    from vescale_amd.pipe import parse_huggingface_model

    stages = parse_huggingface_model(hf_llama, 2)
    ids = torch.randint(0, 256, (2, 8))
    out = _run_chain(stages, ids)
    out.float().pow(2).mean().backward()
    for i, s in enumerate(stages):
        params = list(s.parameters())
        assert params, f"stage {i} has no params"
        assert all(p.grad is not None for p in params), f"stage {i} missing grads"


def test_trace_leaves_model_unpatched(hf_llama):
    """The trace-time identity patches on transformers' masking helpers must
    be fully restored afterwards (eager use of the model keeps working)."""
    import transformers.masking_utils as mu

    before = mu.create_causal_mask
    from vescale_amd.pipe import hf_symbolic_trace

    hf_symbolic_trace(hf_llama)
    assert mu.create_causal_mask is before
    ids = torch.randint(0, 256, (1, 4))
    hf_llama(input_ids=ids)  # still runs eagerly
