"""HF-aware pipeline tracer + PARAMETERS split
(vescale_amd/pipe/pipe_parser.py: hf_symbolic_trace / parse_huggingface_model).

Reference capability: legacy/vescale/pipe/tracer.py:93,626 (hf_symbolic_trace
over HFTracer) + pipe_parser.py:146 (PipelineSplitMethodType.PARAMETERS).
transformers >= 5 removed utils.fx, so ours is a native tracer; these tests
run it against a REAL `transformers` LlamaForCausalLM (random init, no
network) — kwargs forward with a **kwargs catch-all, ModelOutput returns,
data-dependent mask helpers.

FORK SAFETY (tests/README): importing transformers + running fx warms
thread pools that turn later fork+gloo tests into silent hangs, so every
check body runs in a SUBPROCESS (the repo's established isolation pattern,
cf. test_fsdp._llama70b_plan_check).
"""
import os
import subprocess
import sys

import pytest


def _run_in_subprocess(fn_name: str):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = (
        "import sys; sys.path.insert(0, %r); "
        "from tests.test_hf_tracer import %s; %s()"
    ) % (root, fn_name, fn_name)
    out = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=300
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "CHECK_OK" in out.stdout


def _mk_model():
    import torch
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


def _check_trace_parity():
    import torch

    from vescale_amd.pipe import hf_symbolic_trace

    m = _mk_model()
    gm = hf_symbolic_trace(m)
    ids = torch.randint(0, 256, (2, 8))
    ref = m(input_ids=ids).logits
    out = gm(ids)
    out = out[0] if isinstance(out, (tuple, list)) else out
    assert torch.allclose(out, ref, atol=1e-6)
    print("CHECK_OK")


def _check_split_parity():
    import torch

    from vescale_amd.pipe import parse_huggingface_model

    m = _mk_model()
    ids = torch.randint(0, 256, (2, 8))
    ref = m(input_ids=ids).logits
    for num_stages in (2, 3, 4):
        stages = parse_huggingface_model(m, num_stages)
        assert len(stages) == num_stages
        out = _run_chain(stages, ids)
        assert torch.allclose(out, ref, atol=1e-6), num_stages
        # PARAMETERS criterion: no stage empty, balance within 2.5x
        sizes = [sum(p.numel() for p in s.parameters()) for s in stages]
        assert all(sz > 0 for sz in sizes)
        assert max(sizes) / min(sizes) < 2.5, sizes
    print("CHECK_OK")


def _check_backward_flows():
    import torch

    from vescale_amd.pipe import parse_huggingface_model

    m = _mk_model()
    stages = parse_huggingface_model(m, 2)
    ids = torch.randint(0, 256, (2, 8))
    out = _run_chain(stages, ids)
    out.float().pow(2).mean().backward()
    for i, s in enumerate(stages):
        params = list(s.parameters())
        assert params, f"stage {i} has no params"
        assert all(p.grad is not None for p in params), f"stage {i} missing grads"
    print("CHECK_OK")


def _check_patches_restored():
    """The trace-time identity patches on transformers' masking helpers must
    be fully restored afterwards (eager use of the model keeps working)."""
    import torch
    import transformers.masking_utils as mu

    from vescale_amd.pipe import hf_symbolic_trace

    m = _mk_model()
    before = mu.create_causal_mask
    hf_symbolic_trace(m)
    assert mu.create_causal_mask is before
    ids = torch.randint(0, 256, (1, 4))
    m(input_ids=ids)  # still runs eagerly
    print("CHECK_OK")


def test_hf_trace_parity():
    pytest.importorskip("transformers")
    _run_in_subprocess("_check_trace_parity")


def test_hf_parameters_split_parity():
    pytest.importorskip("transformers")
    _run_in_subprocess("_check_split_parity")


def test_hf_split_backward_flows():
    pytest.importorskip("transformers")
    _run_in_subprocess("_check_backward_flows")


def _check_attention_mask_input():
    import torch

    from vescale_amd.pipe import hf_symbolic_trace, parse_huggingface_model

    m = _mk_model()
    ids = torch.randint(0, 256, (2, 8))
    mask = torch.ones(2, 8, dtype=torch.long)
    mask[:, :2] = 0  # left padding
    ref = m(input_ids=ids, attention_mask=mask).logits
    gm = hf_symbolic_trace(m, input_names=("input_ids", "attention_mask"))
    out = gm(ids, mask)
    out = out[0] if isinstance(out, (tuple, list)) else out
    assert torch.allclose(out, ref, atol=1e-6)
    # regression: the config flowing into the mask builder must be the REAL
    # config object (a GraphModule get_attr copy loses _attn_implementation
    # and the builder silently returned no mask)
    stages = parse_huggingface_model(m, 2, input_names=("input_ids", "attention_mask"))
    cur = (ids, mask)
    for s in stages:
        cur = s(*cur)
        if not isinstance(cur, (tuple, list)):
            cur = (cur,)
    assert torch.allclose(cur[0], ref, atol=1e-6)
    print("CHECK_OK")


def test_hf_attention_mask_input():
    pytest.importorskip("transformers")
    _run_in_subprocess("_check_attention_mask_input")


def test_trace_leaves_model_unpatched():
    pytest.importorskip("transformers")
    _run_in_subprocess("_check_patches_restored")
