"""DModule TP/SP tests (CPU/gloo ws=2): forward/grad parity vs single
device — the reference's golden-curve correctness bar."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from tests.common import spawn

from vescale_amd.dtensor import DTensor, Replicate, Shard, init_device_mesh
from vescale_amd.dmodule import parallelize_module


class MLP(nn.Module):
    def __init__(self, d=16):
        super().__init__()
        self.fc1 = nn.Linear(d, 4 * d, bias=False)
        self.fc2 = nn.Linear(4 * d, d, bias=False)

    def forward(self, x):
        return self.fc2(F.relu(self.fc1(x)))


MLP_PLAN = {
    "parameter": {
        r"fc1.weight": [Shard(0)],
        r"fc2.weight": [Shard(1)],
    },
    "forward": {
        "input": [[Replicate()]],          # root input replicate
        r"fc2.output": [[Replicate()]],  # allreduce partial at the end
    },
}


def _t_mlp_tp(rank, ws):
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("TP",))
    torch.manual_seed(0)
    ref = MLP()
    x = torch.randn(8, 16)
    ref_out = ref(x)
    ref_loss = ref_out.pow(2).sum()
    ref_loss.backward()

    torch.manual_seed(0)
    model = MLP()
    parallelize_module(model, mesh, MLP_PLAN)
    assert isinstance(model.fc1.weight.data, DTensor)
    out = model(x)
    assert isinstance(out, DTensor)
    assert torch.allclose(out.to_local(), ref_out, atol=1e-5)
    loss = out.pow(2).sum()
    loss.backward()
    model.finish_grad_sync()
    g1 = model.fc1.weight.grad
    assert torch.allclose(g1.full_tensor(), ref.fc1.weight.grad, atol=1e-4)
    g2 = model.fc2.weight.grad
    assert torch.allclose(g2.full_tensor(), ref.fc2.weight.grad, atol=1e-4)


def test_mlp_tp():
    spawn(2, _t_mlp_tp)


def _t_nanogpt_tp_sp(rank, ws):
    from vescale_amd.models.nanogpt import GPT, gpt_tiny
    from vescale_amd.models.nanogpt_plan import nanogpt_tp_plan

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("TP",))
    torch.manual_seed(0)
    cfg = gpt_tiny()
    ref = GPT(cfg)
    x = torch.randint(0, cfg.vocab_size, (2, 32))
    y = torch.randint(0, cfg.vocab_size, (2, 32))
    _, ref_loss = ref(x, y)
    ref_loss.backward()

    torch.manual_seed(0)
    model = GPT(cfg)
    parallelize_module(model, mesh, nanogpt_tp_plan(sp=True))
    logits, loss = model(x, y)
    lv = loss
    if isinstance(lv, DTensor):
        lv = lv.redistribute(placements=[Replicate()]).to_local()
    assert torch.allclose(lv, ref_loss.detach(), atol=2e-4), (float(lv), float(ref_loss))
    loss.backward()
    model.finish_grad_sync()
    # grad parity on a colwise, a rowwise, and a replicated (LayerNorm) param
    pairs = [
        ("transformer.h.0.attn.c_attn.weight", None),
        ("transformer.h.0.mlp.c_proj.weight", None),
        ("transformer.h.1.ln_1.weight", None),
        ("transformer.wte.weight", None),  # tied with lm_head
    ]
    named_ref = dict(ref.named_parameters())
    named_tp = dict(model.named_parameters())
    for name, _ in pairs:
        rg = named_ref[name].grad
        tg = named_tp[name].grad
        assert tg is not None, name
        if isinstance(tg, DTensor):
            assert not any(p.is_partial() for p in tg.placements), (name, tg.placements)
            tg = tg.full_tensor()
        assert torch.allclose(tg, rg, atol=5e-4), (name, (tg - rg).abs().max())


def test_nanogpt_tp_sp():
    spawn(2, _t_nanogpt_tp_sp)


def _t_factory_mode(rank, ws):
    from vescale_amd.dmodule._factory import FactoryDispatchMode
    from vescale_amd.dtensor import DTensor, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))
    with FactoryDispatchMode(mesh):
        z = torch.zeros(4, 4)
        o = torch.ones(3)
    assert isinstance(z, DTensor) and isinstance(o, DTensor)
    assert torch.equal(z.full_tensor(), torch.zeros(4, 4))
    assert torch.equal(o.full_tensor(), torch.ones(3))


def test_factory_mode():
    spawn(2, _t_factory_mode)


def _t_obj_return(rank, ws):
    """DModules returning non-tensor structures (dicts / dataclass-like)
    pass through the output hooks untouched except tensor leaves
    (reference dmodule/test_obj_return.py)."""
    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import init_device_mesh

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 8)

        def forward(self, x):
            h = self.fc(x)
            return {"hidden": h, "meta": {"n": 3}, "both": (h, "tag")}

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(2)
    net = Net()
    torch.manual_seed(2)
    ref = Net()
    plan = {
        "parameter": {r"fc.weight": [Shard(0)], r"fc.bias": [Shard(0)]},
        "forward": {"input": [[Replicate()]], r"fc.output": [[Replicate()]]},
    }
    net = parallelize_module(net, mesh, plan)
    x = torch.randn(4, 8)
    out = net(x)
    assert out["meta"] == {"n": 3}
    assert out["both"][1] == "tag"
    h = out["hidden"]
    h = h.to_local() if hasattr(h, "to_local") else h
    assert torch.allclose(h, ref(x)["hidden"], atol=1e-6)


def test_obj_return():
    spawn(2, _t_obj_return)


def _t_obj_return_named(rank, ws):
    """Dict-shaped plans: name-keyed input plans bind forward() params
    (positional or kw), and name-keyed output plans convert fields of
    dataclass / Mapping returns (reference dmodule/test_obj_return.py +
    _hook.py dict-like paths)."""
    import dataclasses

    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import init_device_mesh

    @dataclasses.dataclass
    class Out:
        last_hidden: torch.Tensor = None
        hidden_plus: torch.Tensor = None
        note: str = "keep"

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 8)

        def forward(self, x, scale=1.0):
            h = self.fc(x) * scale
            return Out(last_hidden=h, hidden_plus=h + 1)

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(3)
    net = Net()
    torch.manual_seed(3)
    ref = Net()
    plan = {
        "parameter": {},
        "forward": {
            "input": {"x": [Replicate()]},
            "output": {"last_hidden": [Replicate()], "hidden_plus": [Shard(0)]},
        },
    }
    net = parallelize_module(net, mesh, plan)
    x = torch.randn(4, 8)
    # kwargs path exercises the signature binding too
    out = net(x, scale=2.0)
    assert isinstance(out.last_hidden, DTensor)
    assert tuple(out.last_hidden.placements) == (Replicate(),)
    assert isinstance(out.hidden_plus, DTensor)
    assert tuple(out.hidden_plus.placements) == (Shard(0),)
    assert out.note == "keep"
    want = ref(x, scale=2.0)
    assert torch.allclose(out.last_hidden.to_local(), want.last_hidden, atol=1e-6)
    assert torch.allclose(out.hidden_plus.full_tensor(), want.hidden_plus, atol=1e-6)

    # Mapping return + dict plan
    class NetD(nn.Module):
        def __init__(self):
            super().__init__()
            self.fc = nn.Linear(8, 8)

        def forward(self, x):
            h = self.fc(x)
            return {"hidden": h, "tag": "t"}

    torch.manual_seed(4)
    netd = parallelize_module(
        NetD(), mesh,
        {"parameter": {}, "forward": {"output": {"hidden": [Replicate()]}}},
    )
    outd = netd(x)
    assert isinstance(outd["hidden"], DTensor) and outd["tag"] == "t"


def test_obj_return_named():
    spawn(2, _t_obj_return_named)


def _t_state_dict_saveload(rank, ws):
    """torch.save/load roundtrip of a parallelized module's state dict
    (reference dmodule/test_saveload.py): entries are DTensors, reload
    restores bitwise-equal locals."""
    import tempfile

    from vescale_amd.dtensor import init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(9)
    net = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 8))
    plan = {
        "parameter": {
            r"0.weight": [Shard(0)], r"0.bias": [Shard(0)],
            r"2.weight": [Shard(1)], r"2.bias": [Replicate()],
        },
        "forward": {"input": [[Replicate()]]},
    }
    net = parallelize_module(net, mesh, plan)
    sd = net.state_dict()
    assert all(isinstance(v, DTensor) for v in sd.values()), sd.keys()
    path = tempfile.mktemp(suffix=f".r{rank}.pt")
    torch.save(sd, path)
    # perturb, then restore
    with torch.no_grad():
        for p in net.parameters():
            p.add_(1.0)
    loaded = torch.load(path, weights_only=False)
    net.load_state_dict(loaded)
    for k, v in net.state_dict().items():
        assert torch.equal(v._local_tensor, sd[k]._local_tensor), k
    import os as _os
    _os.unlink(path)


def test_state_dict_saveload():
    spawn(2, _t_state_dict_saveload)


def _t_grad_placement_enforce(rank, ws):
    """PlacementsInterface(grad=...) forces the parameter's gradient into
    a chosen placement via a hook (reference PostHookGrad): here a
    row-parallel weight's naturally-Partial grad is reduced to Replicate
    at backward time instead of at finish_grad_sync."""
    from vescale_amd.dmodule import PlacementsInterface as PI
    from vescale_amd.dtensor import Partial, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(6)
    net = nn.Linear(8, 8, bias=False)
    torch.manual_seed(6)
    ref = nn.Linear(8, 8, bias=False)
    plan = {
        "parameter": {r"weight": PI([Shard(1)], grad=[Replicate()])},
        "forward": {"input": [[Replicate()]], "output": [[Replicate()]]},
    }
    net = parallelize_module(net, mesh, plan)
    x = torch.randn(4, 8, generator=torch.Generator().manual_seed(7))
    out = net(x)
    (out.to_local() if isinstance(out, DTensor) else out).pow(2).mean().backward()
    g = net.weight.grad
    assert isinstance(g, DTensor) and g.placements[0].is_replicate(), g.placements
    ref(x).pow(2).mean().backward()
    assert torch.allclose(g.full_tensor(), ref.weight.grad, atol=1e-6)


def test_grad_placement_enforce():
    spawn(2, _t_grad_placement_enforce)


def _t_dict_input_plan_binding(rank, ws):
    """Dict input plans bind by PARAMETER NAME across calling conventions
    (reference test_fwd_plan.py dict_fwd_plan): positional, keyword,
    kw-only, and default-valued args all convert."""
    from vescale_amd.dtensor import init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))

    class Pos(nn.Module):
        def forward(self, a, b=None):
            return a if b is None else a + b

    class KwOnly(nn.Module):
        def forward(self, *, a):
            return a

    plan = {"parameter": {}, "forward": {"input": {"a": [Shard(0)]}}}
    a = torch.ones(2 * ws, 2)

    m = parallelize_module(Pos(), mesh, dict(plan))
    out = m(a)                      # positional
    assert isinstance(out, DTensor) and out.placements[0].is_shard(0)
    out = m(a=a)                    # keyword
    assert isinstance(out, DTensor) and out.placements[0].is_shard(0)
    out = m(a, b=torch.zeros(2 * ws, 2))  # unplanned arg passes through (replicate)
    assert isinstance(out, DTensor)

    m2 = parallelize_module(KwOnly(), mesh, dict(plan))
    out = m2(a=a)                   # kw-only binding
    assert isinstance(out, DTensor) and out.placements[0].is_shard(0)


def test_dict_input_plan_binding():
    spawn(2, _t_dict_input_plan_binding)


def _t_ulysses_sdpa(rank, ws):
    """Ulysses SP attention: numerics + grads match single-device SDPA;
    exactly 4 all-to-alls per forward (one per q/k/v/out)."""
    from vescale_amd.debug import CommDebugMode
    from vescale_amd.dmodule.ulysses import ulysses_sdpa
    from vescale_amd.dtensor import distribute_tensor, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))
    B, H, S, D = 2, 4, 8, 16
    g = torch.Generator().manual_seed(3)
    qg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    kg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    vg = torch.randn(B, H, S, D, generator=g, requires_grad=True)

    for causal in (False, True):
        q = distribute_tensor(qg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        k = distribute_tensor(kg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        v = distribute_tensor(vg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        with CommDebugMode() as cm:
            out = ulysses_sdpa(q, k, v, is_causal=causal)
        assert cm.get_comm_counts().get("mesh_all_to_all", 0) == 4, cm.get_comm_counts()
        assert out.placements[0].is_shard(2)

        import torch.nn.functional as F
        ref = F.scaled_dot_product_attention(qg, kg, vg, is_causal=causal)
        assert torch.allclose(out.full_tensor(), ref, atol=1e-5)

        # grads flow back through the four transposed all-to-alls
        out.to_local().pow(2).mean().backward()
        (ref.pow(2).mean() / 1.0).backward(retain_graph=False)
        gq = q.grad.full_tensor() if isinstance(q.grad, DTensor) else q.grad
        # loss differs by the mean over the local vs global numel: local
        # mean over 1/ws of elements scales grads by ws
        assert torch.allclose(gq / ws, qg.grad / 1.0, atol=1e-5), causal
        qg.grad = kg.grad = vg.grad = None


def test_ulysses_sdpa():
    spawn(2, _t_ulysses_sdpa)


def _t_ring_sdpa(rank, ws):
    """Ring attention: exact numerics + grads vs single-device SDPA,
    causal and non-causal, incl. GQA."""
    from vescale_amd.dmodule.ring_attention import ring_sdpa
    from vescale_amd.dtensor import distribute_tensor, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,))
    B, H, S, D = 2, 4, 8, 16
    g = torch.Generator().manual_seed(4)
    qg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    kg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    vg = torch.randn(B, H, S, D, generator=g, requires_grad=True)

    import torch.nn.functional as F

    for causal in (False, True):
        q = distribute_tensor(qg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        k = distribute_tensor(kg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        v = distribute_tensor(vg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
        out = ring_sdpa(q, k, v, is_causal=causal)
        assert out.placements[0].is_shard(2)
        ref = F.scaled_dot_product_attention(qg, kg, vg, is_causal=causal)
        assert torch.allclose(out.full_tensor(), ref, atol=1e-5), causal

        out.to_local().pow(2).sum().backward()
        ref.pow(2).sum().backward()
        for dt, pl in ((q, qg), (k, kg), (v, vg)):
            gd = dt.grad.full_tensor() if isinstance(dt.grad, DTensor) else dt.grad
            assert torch.allclose(gd, pl.grad, atol=1e-5), causal
        qg.grad = kg.grad = vg.grad = None

    # GQA: 2 kv heads under 4 q heads
    kq = torch.randn(B, 2, S, D, generator=g)
    vq = torch.randn(B, 2, S, D, generator=g)
    q = distribute_tensor(qg.detach().clone(), mesh, [Shard(2)])
    k2 = distribute_tensor(kq, mesh, [Shard(2)])
    v2 = distribute_tensor(vq, mesh, [Shard(2)])
    out = ring_sdpa(q, k2, v2, is_causal=True)
    ref = F.scaled_dot_product_attention(
        qg.detach(), kq.repeat_interleave(2, 1), vq.repeat_interleave(2, 1), is_causal=True
    )
    assert torch.allclose(out.full_tensor(), ref, atol=1e-5)


def test_ring_sdpa():
    spawn(2, _t_ring_sdpa)


def _t_ring_sdpa_ws4(rank, ws):
    """Ring attention at p=4 (multi-hop ring, mixed masked/unmasked)."""
    from vescale_amd.dmodule.ring_attention import ring_sdpa
    from vescale_amd.dtensor import distribute_tensor, init_device_mesh
    import torch.nn.functional as F

    mesh = init_device_mesh("cpu", (ws,))
    B, H, S, D = 1, 2, 16, 8
    g = torch.Generator().manual_seed(5)
    qg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    kg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    vg = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    q = distribute_tensor(qg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
    k = distribute_tensor(kg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
    v = distribute_tensor(vg.detach().clone().requires_grad_(True), mesh, [Shard(2)])
    out = ring_sdpa(q, k, v, is_causal=True)
    ref = F.scaled_dot_product_attention(qg, kg, vg, is_causal=True)
    assert torch.allclose(out.full_tensor(), ref, atol=1e-5)
    out.to_local().pow(2).sum().backward()
    ref.pow(2).sum().backward()
    gq = q.grad.full_tensor() if hasattr(q.grad, "full_tensor") else q.grad
    assert torch.allclose(gq, qg.grad, atol=1e-5)


def test_ring_sdpa_ws4():
    spawn(4, _t_ring_sdpa_ws4)


def _t_sp_on_submesh(rank, ws):
    """Ring + Ulysses over the SP dim of a 2-D (DP, SP) mesh: subgroup
    global-rank mapping in the ring shift, per-DP-group independent data."""
    from vescale_amd.dmodule.ring_attention import ring_sdpa
    from vescale_amd.dmodule.ulysses import ulysses_sdpa
    from vescale_amd.dtensor import distribute_tensor, init_device_mesh
    import torch.nn.functional as F

    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("DP", "SP"))
    sp = mesh["SP"]
    dp_idx = mesh.get_coordinate()[0]
    B, H, S, D = 1, 2, 8, 4
    g = torch.Generator().manual_seed(100 + dp_idx)  # different per DP group
    qg = torch.randn(B, H, S, D, generator=g)
    kg = torch.randn(B, H, S, D, generator=g)
    vg = torch.randn(B, H, S, D, generator=g)
    q = distribute_tensor(qg, sp, [Shard(2)])
    k = distribute_tensor(kg, sp, [Shard(2)])
    v = distribute_tensor(vg, sp, [Shard(2)])
    ref = F.scaled_dot_product_attention(qg, kg, vg, is_causal=True)
    out_r = ring_sdpa(q, k, v, is_causal=True)
    assert torch.allclose(out_r.full_tensor(), ref, atol=1e-5)
    out_u = ulysses_sdpa(q, k, v, is_causal=True)
    assert torch.allclose(out_u.full_tensor(), ref, atol=1e-5)


def test_sp_on_submesh():
    spawn(4, _t_sp_on_submesh)


def _t_sp_shape_sweep(rank, ws):
    """Ring + Ulysses across diverse (B,H,S,D,causal,gqa) shapes."""
    from vescale_amd.dmodule.ring_attention import ring_sdpa
    from vescale_amd.dmodule.ulysses import ulysses_sdpa
    from vescale_amd.dtensor import distribute_tensor, init_device_mesh
    import torch.nn.functional as F

    mesh = init_device_mesh("cpu", (ws,))
    cases = [
        (1, 2, 4, 8, False, 1),
        (3, 4, 8, 16, True, 1),
        (2, 8, 16, 4, True, 2),    # GQA 8q/4kv
        (1, 2, 32, 32, False, 2),  # GQA 2q/1kv
        (2, 6, 12, 8, True, 3),    # GQA 6q/2kv, S not power of 2
    ]
    for idx, (B, H, S, D, causal, gqa) in enumerate(cases):
        if H % ws or S % ws or (H // gqa) % ws:
            continue
        g = torch.Generator().manual_seed(50 + idx)
        qg = torch.randn(B, H, S, D, generator=g)
        kg = torch.randn(B, H // gqa, S, D, generator=g)
        vg = torch.randn(B, H // gqa, S, D, generator=g)
        ref = F.scaled_dot_product_attention(
            qg, kg.repeat_interleave(gqa, 1), vg.repeat_interleave(gqa, 1),
            is_causal=causal,
        )
        q = distribute_tensor(qg, mesh, [Shard(2)])
        k = distribute_tensor(kg, mesh, [Shard(2)])
        v = distribute_tensor(vg, mesh, [Shard(2)])
        out_r = ring_sdpa(q, k, v, is_causal=causal)
        assert torch.allclose(out_r.full_tensor(), ref, atol=1e-5), (idx, "ring")
        if (H // gqa) % ws == 0 and H % ws == 0:
            out_u = ulysses_sdpa(q, k, v, is_causal=causal)
            assert torch.allclose(out_u.full_tensor(), ref, atol=1e-5), (idx, "uly")


def test_sp_shape_sweep():
    spawn(2, _t_sp_shape_sweep)
