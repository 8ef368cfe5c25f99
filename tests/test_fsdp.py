"""FSDP engine tests (CPU/gloo): ws=1 vs ws=2 loss-curve parity — the
reference's golden-curve methodology on the tiny Llama."""
import json
import os
import tempfile

import pytest
import torch

from tests.common import spawn

from vescale_amd.fsdp import FSDP, FlatAdamW
from vescale_amd.models.llama import LlamaModel, llama_tiny


def _make_batches(cfg, n_steps, batch=4, seq=32):
    g = torch.Generator().manual_seed(1234)
    out = []
    for _ in range(n_steps):
        x = torch.randint(0, cfg.vocab_size, (batch, seq), generator=g)
        y = torch.roll(x, -1, dims=1)
        out.append((x, y))
    return out


def _train(rank, ws, out_path, n_steps):
    from vescale_amd.dtensor import init_device_mesh

    torch.manual_seed(42)
    cfg = llama_tiny()
    model = LlamaModel(cfg)
    model.init_weights()
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.float32, device=torch.device("cpu"))
    opt = FlatAdamW(eng, lr=1e-3, weight_decay=0.0, grad_clip=1.0)
    batches = _make_batches(cfg, n_steps)
    losses = []
    for x, y in batches:
        # data-parallel split of the batch
        xs = torch.chunk(x, ws)[rank]
        ys = torch.chunk(y, ws)[rank]
        loss = eng(xs, ys)
        loss.backward()
        opt.step()
        # global mean loss for comparison
        l = loss.detach().clone()
        if ws > 1:
            torch.distributed.all_reduce(l)
            l /= ws
        losses.append(float(l))
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(losses, f)


@pytest.mark.parametrize("ws", [2, 4])
def test_fsdp_ws_parity(ws):
    """ws=1 vs ws=N loss parity — N=4 exercises multi-unit RaggedShard
    math beyond the pairwise case (insurance for the 8-GPU scale run)."""
    n_steps = 4
    with tempfile.TemporaryDirectory() as td:
        p1 = os.path.join(td, "ws1.json")
        p2 = os.path.join(td, "wsN.json")
        spawn(1, _train, p1, n_steps)
        spawn(ws, _train, p2, n_steps)
        l1 = json.load(open(p1))
        l2 = json.load(open(p2))
        assert len(l1) == len(l2) == n_steps
        for a, b in zip(l1, l2):
            assert abs(a - b) < 2e-3, f"loss diverged: {l1} vs {l2}"
        # loss must actually decrease
        assert l1[-1] < l1[0]


def _t_state_dict(rank, ws):
    from vescale_amd.dtensor import init_device_mesh

    torch.manual_seed(42)
    cfg = llama_tiny()
    model = LlamaModel(cfg)
    model.init_weights()
    ref = {k: v.detach().clone() for k, v in model.named_parameters()}
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.float32, device=torch.device("cpu"))
    sd = eng.sharded_state_dict()
    assert len(sd) == len(ref)
    for k, d in sd.items():
        full = d.full_tensor()
        assert torch.allclose(full, ref[k].reshape(-1), atol=1e-6), k


def test_fsdp_sharded_state_dict():
    spawn(2, _t_state_dict)


def _t_grad_accumulation(rank, ws):
    """Two micro-backwards before step must ACCUMULATE (not overwrite)."""
    from vescale_amd.dtensor import init_device_mesh

    torch.manual_seed(7)
    cfg = llama_tiny()
    model = LlamaModel(cfg)
    model.init_weights()
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.float32, device=torch.device("cpu"))
    g = torch.Generator().manual_seed(3)
    x1 = torch.randint(0, cfg.vocab_size, (2, 32), generator=g)
    x2 = torch.randint(0, cfg.vocab_size, (2, 32), generator=g)

    # accumulate two backwards
    eng(x1, torch.roll(x1, -1, 1)).backward()
    eng(x2, torch.roll(x2, -1, 1)).backward()
    eng.finish_grad_sync()
    acc = {u.name: u.grad_shard.clone() for u in eng.units}

    # reference: separate single-backward runs summed
    eng.zero_grad_buffers()
    eng(x1, torch.roll(x1, -1, 1)).backward()
    eng.finish_grad_sync()
    g1 = {u.name: u.grad_shard.clone() for u in eng.units}
    eng.zero_grad_buffers()
    eng(x2, torch.roll(x2, -1, 1)).backward()
    eng.finish_grad_sync()
    for u in eng.units:
        want = g1[u.name] + u.grad_shard
        assert torch.allclose(acc[u.name], want, atol=1e-5), u.name


def test_fsdp_grad_accumulation():
    spawn(2, _t_grad_accumulation)


def _t_ws8_step(rank, ws):
    """One step at ws=8 (the scale-run world size) — shard math, comms
    and optimizer must hold beyond the small worlds."""
    from vescale_amd.dtensor import init_device_mesh

    torch.manual_seed(42)
    cfg = llama_tiny()
    model = LlamaModel(cfg)
    model.init_weights()
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.float32, device=torch.device("cpu"))
    opt = FlatAdamW(eng, lr=1e-3, grad_clip=1.0)
    g = torch.Generator().manual_seed(5)
    x = torch.randint(0, cfg.vocab_size, (8, 32), generator=g)
    xs = torch.chunk(x, ws)[rank]
    loss = eng(xs, torch.roll(xs, -1, 1))
    loss.backward()
    opt.step()
    l2 = eng(xs, torch.roll(xs, -1, 1))
    assert float(l2) < float(loss), "loss should drop after one step"


def test_fsdp_ws8_step():
    spawn(8, _t_ws8_step)


def _llama70b_plan_check():
    from vescale_amd.initialize import deferred_init, is_deferred
    from vescale_amd.models.llama import LlamaModel, llama3_70b

    cfg = llama3_70b()
    model = deferred_init(LlamaModel, cfg)
    assert is_deferred(model)
    n_params = sum(p.numel() for p in model.parameters())
    assert 68e9 < n_params < 72e9, n_params

    ws = 8
    GB = 1024**3
    param_shard = n_params * 2 / ws            # bf16 shard
    grad_shard = n_params * 2 / ws             # bf16 grad shard (ZeRO-2+)
    opt_state = n_params * 12 / ws             # fp32 master + m + v
    # transient: largest unit gathered full (embeddings+head ~= 2*vocab*d)
    full_unit = 2 * cfg.vocab_size * cfg.dim * 2
    # activations, bs1 x seq8192, full recompute granularity: per-layer
    # boundary activation + one layer live
    act = 8192 * cfg.dim * 2 * (cfg.n_layers + 6)
    total = param_shard + grad_shard + opt_state + full_unit + act
    assert total < 288 * GB, f"plan needs {total/GB:.0f} GB"
    # and it does NOT fit a 141 GB (H200-class) device -> the 288 GB HBM
    # is what makes single-node 70B FSDP8 viable
    assert total > 141 * GB
    print("PLAN_OK")


def test_llama70b_memory_plan_fits_mi355x():
    """BASELINE config 'Llama-3 70B FSDP+TP 2D mesh on 8x MI355X':
    construct the 70B architecture on the meta device (deferred init, no
    allocation) and check the per-GPU steady-state memory plan fits the
    288 GB HBM3E budget with FSDP=8 sharding of params/grads/opt state.
    Runs in a SUBPROCESS: building even a meta 70B module warms torch
    thread pools, and later fork-based gloo tests would inherit a locked
    pool and hang."""
    import subprocess
    import sys

    code = (
        "import sys; sys.path.insert(0, %r); "
        "from tests.test_fsdp import _llama70b_plan_check; "
        "_llama70b_plan_check()"
    ) % os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True,
        timeout=300,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    assert "PLAN_OK" in out.stdout


def _t_frozen_param(rank, ws):
    """Regression: a unit containing an unused requires_grad
    param must still reduce-scatter the grads it DID produce, and the
    readiness state must not leak into the next step."""
    import torch.nn as nn
    from vescale_amd.dtensor import init_device_mesh

    torch.manual_seed(7)

    class Branchy(nn.Module):
        def __init__(self):
            super().__init__()
            self.used = nn.Linear(8, 8, bias=False)
            self.unused = nn.Linear(8, 8, bias=False)  # requires_grad, no grad

        def forward(self, x):
            return self.used(x)

    model = Branchy()
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    eng = FSDP(model, mesh, param_dtype=torch.float32,
               device=torch.device("cpu"))
    opt = FlatAdamW(eng, lr=1e-2, weight_decay=0.0)
    g = torch.Generator().manual_seed(5)
    for step in range(3):
        used_before = model.used.weight.detach().clone()
        unused_before = model.unused.weight.detach().clone()
        x = torch.randn(4, 8, generator=g)
        out = eng(torch.chunk(x, ws)[rank])
        loss = out.pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        # the used param's grads must be reduced + applied EVERY step (the
        # old counter wedged the unit after step 1 -> no update), and the
        # zero-grad param must not drift
        assert not torch.allclose(model.used.weight, used_before), step
        assert torch.allclose(model.unused.weight, unused_before), step


def test_fsdp_unused_param_unit_flush():
    spawn(2, _t_frozen_param)
