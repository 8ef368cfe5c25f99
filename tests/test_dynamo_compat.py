"""torch.compile over DModule TP models — eager-first dynamo compat.

Reference patch #9 (patched_pytorch_v2.2.1_rc3.patch:996-1378) taught
dynamo to trace its DTensor.  Our eager-first equivalent: DTensor
construction and the plan hooks are OPAQUE to dynamo (graph breaks), so
torch.compile compiles the local compute between DTensor boundaries and
the redistributes run eagerly.  These tests pin that contract: compiled
forward is bit-identical to eager, backward produces grads, and a
recompile-free second step works.
"""
import pytest
import torch
import torch.nn as nn

from tests.common import spawn

from vescale_amd import Replicate, Shard, init_device_mesh


def _t_compile_tp(rank, ws):
    from vescale_amd.dmodule import parallelize_module

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(3)
    m = nn.Sequential(nn.Linear(16, 16), nn.Tanh(), nn.Linear(16, 16))
    torch.manual_seed(3)
    ref = nn.Sequential(nn.Linear(16, 16), nn.Tanh(), nn.Linear(16, 16))
    plan = {
        "parameter": {
            r"0.weight": [Shard(0)],
            r"0.bias": [Shard(0)],
            r"2.weight": [Shard(1)],
            r"2.bias": [Replicate()],
        },
        "forward": {"input": [[Replicate()]], "output": [[Replicate()]]},
    }
    m = parallelize_module(m, mesh, plan)
    cm = torch.compile(m, backend="eager", fullgraph=False)
    torch.manual_seed(9)
    for step in range(2):  # second step: no recompile-induced failure
        x = torch.randn(4, 16)
        out = cm(x)
        out_l = out.to_local() if hasattr(out, "to_local") else out
        assert torch.equal(out_l, ref(x)), step
        out_l.pow(2).mean().backward()
        ref(x).pow(2).mean().backward()
    for p in m.parameters():
        assert p.grad is not None


def test_compile_tp_dmodule():
    spawn(2, _t_compile_tp)


def test_compile_plain_dtensor_ops():
    """compile a function mixing plain and DTensor math (single rank)."""

    def body(rank, ws):
        from vescale_amd import distribute_tensor

        mesh = init_device_mesh("cpu", (ws,))
        d = distribute_tensor(torch.arange(8, dtype=torch.float32), mesh, [Shard(0)])

        def fn(t, scalar):
            return (t * scalar + 1.0).relu()

        cfn = torch.compile(fn, backend="eager", fullgraph=False)
        out = cfn(d, 2.0)
        ref = fn(d.full_tensor(), 2.0)
        assert torch.equal(out.full_tensor(), ref)

    spawn(2, body)
