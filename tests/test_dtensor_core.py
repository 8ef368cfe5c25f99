"""DTensor core tests on CPU/gloo world_size=2.

Mirrors the reference test pyramid tier 1: distribute /
redistribute / op parity vs a single-device reference.
"""
import pytest
import torch
import torch.nn.functional as F

from tests.common import spawn

from vescale_amd import (
    DTensor,
    InterleavedShard,
    Partial,
    RaggedShard,
    Replicate,
    Shard,
    distribute_tensor,
    init_device_mesh,
)


# ---------------------------------------------------------------------------
# pure placement math (no comm)
# ---------------------------------------------------------------------------
def test_shard_chunk_math():
    assert Shard.chunk_size(10, 4, 0) == 3
    assert Shard.chunk_size(10, 4, 3) == 1
    assert [Shard.chunk_size(10, 4, i) for i in range(4)] == [3, 3, 3, 1]
    assert [Shard.chunk_offset(10, 4, i) for i in range(4)] == [0, 3, 6, 9]
    # chunk semantics match torch.chunk
    t = torch.arange(10)
    chunks = Shard(0).split_tensor(t, 4)
    ref = list(torch.chunk(t, 4))
    for c, r in zip(chunks, ref):
        assert torch.equal(c, r)


def test_interleaved_shard_split():
    t = torch.arange(12)
    p = InterleavedShard(0, 2)
    chunks = p.split_tensor(t, 3)
    # view as [2,6], shard cols into 3 -> rank0 gets cols 0,1 of each row
    assert torch.equal(chunks[0], torch.tensor([0, 1, 6, 7]))
    assert torch.equal(chunks[2], torch.tensor([4, 5, 10, 11]))


def test_ragged_shard_split():
    t = torch.arange(24).reshape(6, 4)
    p = RaggedShard((0,), (1, 2))  # units over flattened leading dim
    chunks = p.split_tensor(t, 2)
    assert chunks[0].numel() == 8 and chunks[1].numel() == 16
    assert torch.equal(chunks[0], torch.arange(8))
    rec = RaggedShard.reconstruct(torch.cat(chunks), (6, 4))
    assert torch.equal(rec, t)


# ---------------------------------------------------------------------------
# ws=2 distributed behavior
# ---------------------------------------------------------------------------
def _t_distribute(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(5)
    g = torch.randn(9, 4)  # uneven on purpose
    d = distribute_tensor(g, mesh, [Shard(0)])
    assert d.shape == (9, 4)
    expect = torch.chunk(g, ws)[rank] if rank < ws else None
    assert torch.equal(d._local_tensor, expect)
    full = d.full_tensor()
    assert torch.equal(full, g)


def test_distribute_uneven():
    spawn(2, _t_distribute)


def _t_redistribute_matrix(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(7)
    g = torch.randn(8, 6)
    for src in ([Shard(0)], [Shard(1)], [Replicate()]):
        for dst in ([Shard(0)], [Shard(1)], [Replicate()]):
            d = distribute_tensor(g, mesh, src)
            r = d.redistribute(placements=dst)
            assert torch.equal(r.full_tensor(), g), f"{src}->{dst}"
    # Partial -> Replicate / Shard
    local = torch.full((4, 4), float(rank + 1))
    d = DTensor.from_local(local, mesh, [Partial()], shape=torch.Size((4, 4)))
    rep = d.redistribute(placements=[Replicate()])
    assert torch.equal(rep._local_tensor, torch.full((4, 4), 3.0))
    d = DTensor.from_local(local, mesh, [Partial()], shape=torch.Size((4, 4)))
    sh = d.redistribute(placements=[Shard(0)])
    assert sh._local_tensor.shape == (2, 4)
    assert torch.equal(sh.full_tensor(), torch.full((4, 4), 3.0))


def test_redistribute_matrix():
    spawn(2, _t_redistribute_matrix)


def _t_redistribute_uneven(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    g = torch.arange(7 * 3, dtype=torch.float32).reshape(7, 3)
    d = distribute_tensor(g, mesh, [Shard(0)])
    r = d.redistribute(placements=[Replicate()])
    assert torch.equal(r._local_tensor, g)
    r2 = r.redistribute(placements=[Shard(1)])
    assert torch.equal(r2.full_tensor(), g)


def test_redistribute_uneven():
    spawn(2, _t_redistribute_uneven)


def _t_ragged_redistribute(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    g = torch.arange(12, dtype=torch.float32)
    p = RaggedShard((0,), (1, 3))
    d = distribute_tensor(g, mesh, [p])
    assert d._local_tensor.numel() == (3 if rank == 0 else 9)
    # RS -> R
    full = d.full_tensor()
    assert torch.equal(full, g)
    # RS -> RS' (interval-intersection all_to_all)
    p2 = RaggedShard((0,), (2, 2))
    d2 = d.redistribute(placements=[p2])
    assert d2._local_tensor.numel() == 6
    assert torch.equal(d2.full_tensor(), g)


def test_ragged_redistribute():
    spawn(2, _t_ragged_redistribute)


def _t_mm_parity(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(11)
    a = torch.randn(8, 6)
    b = torch.randn(6, 4)
    ref = a @ b
    for pa in ([Shard(0)], [Shard(1)], [Replicate()]):
        for pb in ([Shard(0)], [Shard(1)], [Replicate()]):
            da = distribute_tensor(a, mesh, pa)
            db = distribute_tensor(b, mesh, pb)
            dc = da @ db
            assert torch.allclose(dc.full_tensor(), ref, atol=1e-5), f"{pa}x{pb}"


def test_mm_parity():
    spawn(2, _t_mm_parity)


def _t_linear_training_step(rank, ws):
    """2-layer MLP with Shard(0)/Shard(1) weights (colwise->rowwise TP) —
    the BASELINE.json plumbing config."""
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(3)
    x_g = torch.randn(4, 8)
    w1_g = torch.randn(16, 8)
    w2_g = torch.randn(8, 16)
    # reference
    xr = x_g.clone().requires_grad_(True)
    w1r = w1_g.clone().requires_grad_(True)
    w2r = w2_g.clone().requires_grad_(True)
    loss_r = F.linear(F.relu(F.linear(xr, w1r)), w2r).pow(2).sum()
    loss_r.backward()

    x = distribute_tensor(x_g, mesh, [Replicate()]).requires_grad_(True)
    w1 = distribute_tensor(w1_g, mesh, [Shard(0)]).requires_grad_(True)
    w2 = distribute_tensor(w2_g, mesh, [Shard(1)]).requires_grad_(True)
    h = F.relu(F.linear(x, w1))
    out = F.linear(h, w2)
    loss = out.pow(2).sum()
    lf = loss.redistribute(placements=[Replicate()])
    assert torch.allclose(lf.to_local(), loss_r.detach(), atol=1e-4)
    loss.backward()
    gw1 = w1.grad.full_tensor() if isinstance(w1.grad, DTensor) else w1.grad
    assert torch.allclose(gw1, w1r.grad, atol=1e-4)
    gw2 = w2.grad.full_tensor() if isinstance(w2.grad, DTensor) else w2.grad
    assert torch.allclose(gw2, w2r.grad, atol=1e-4)


def test_linear_training_step():
    spawn(2, _t_linear_training_step)


def _t_reductions(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(13)
    g = torch.randn(8, 6)
    d = distribute_tensor(g, mesh, [Shard(0)])
    checks = [
        (d.sum(), g.sum()),
        (d.mean(), g.mean()),
        (d.sum(dim=0), g.sum(dim=0)),
        (d.sum(dim=1), g.sum(dim=1)),
        (d.amax(), g.amax()),
        (d.pow(2).sum(), g.pow(2).sum()),
    ]
    for got, want in checks:
        gf = got.redistribute(placements=[Replicate()]).to_local()
        assert torch.allclose(gf, want, atol=1e-5)
    n = torch.linalg.vector_norm(d)
    assert torch.allclose(n.to_local(), torch.linalg.vector_norm(g), atol=1e-5)


def test_reductions():
    spawn(2, _t_reductions)


def _t_embedding_vocab_parallel(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(17)
    w = torch.randn(10, 4)
    idx = torch.randint(0, 10, (3, 5))
    ref = F.embedding(idx, w)
    dw = distribute_tensor(w, mesh, [Shard(0)])
    didx = distribute_tensor(idx, mesh, [Replicate()])
    out = F.embedding(didx, dw)
    assert any(p.is_partial() for p in out.placements)
    assert torch.allclose(out.full_tensor(), ref, atol=1e-5)


def test_embedding_vocab_parallel():
    spawn(2, _t_embedding_vocab_parallel)


def _t_cross_entropy_batch_sharded(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(19)
    logits = torch.randn(8, 12)
    tgt = torch.randint(0, 12, (8,))
    ref = F.cross_entropy(logits, tgt)
    dl = distribute_tensor(logits, mesh, [Shard(0)]).requires_grad_(True)
    dt = distribute_tensor(tgt, mesh, [Shard(0)])
    loss = F.cross_entropy(dl, dt)
    lv = loss.redistribute(placements=[Replicate()]).to_local()
    assert torch.allclose(lv, ref, atol=1e-5)
    loss.backward()
    logits_r = logits.clone().requires_grad_(True)
    F.cross_entropy(logits_r, tgt).backward()
    assert torch.allclose(dl.grad.full_tensor(), logits_r.grad, atol=1e-5)


def test_cross_entropy_batch_sharded():
    spawn(2, _t_cross_entropy_batch_sharded)


def _t_2d_mesh(rank, ws):
    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("DP", "TP"))
    torch.manual_seed(23)
    g = torch.randn(8, 8)
    d = distribute_tensor(g, mesh, [Shard(0), Shard(1)])
    coord = mesh.get_coordinate()
    assert d._local_tensor.shape == (4, 4)
    assert torch.equal(d.full_tensor(), g)
    r = d.redistribute(placements=[Replicate(), Shard(0)])
    assert torch.equal(r.full_tensor(), g)


@pytest.mark.skipif(False, reason="")
def test_2d_mesh():
    spawn(4, _t_2d_mesh)


def _t_submesh(rank, ws):
    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("DP", "TP"))
    tp = mesh["TP"]
    assert tp.ndim == 1 and tp.size() == 2
    dp = mesh["DP"]
    assert dp.size() == 2
    g = torch.randn(4, 4)
    d = distribute_tensor(g, tp, [Shard(0)])
    assert torch.equal(d.full_tensor(), g)


def test_submesh():
    spawn(4, _t_submesh)


def _t_softmax_dropout(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(29)
    g = torch.randn(6, 8)
    d = distribute_tensor(g, mesh, [Shard(0)])
    s = torch.softmax(d, dim=-1)
    assert torch.allclose(s.full_tensor(), torch.softmax(g, dim=-1), atol=1e-6)
    # dropout p=0 keeps values
    o = F.dropout(d, p=0.0, training=True)
    assert torch.allclose(o.full_tensor(), g)


def test_softmax_dropout():
    spawn(2, _t_softmax_dropout)


def _t_redistribute_fuzz(rank, ws):
    """Seeded fuzz over shapes x placement pairs: full_tensor is invariant
    under any redistribute chain."""
    import random

    mesh = init_device_mesh("cpu", (ws,))
    rng = random.Random(1234)
    placement_pool = [
        [Replicate()], [Shard(0)], [Shard(1)], [Shard(2)],
        [InterleavedShard(0, 2)],
        [RaggedShard((0,), (1, 2))], [RaggedShard((0, 1), (3, 1))],
    ]
    for trial in range(30):
        dims = rng.choice([2, 3])
        shape = [rng.choice([2, 4, 6, 8, 12]) for _ in range(dims)]
        # keep divisibility for interleave/ragged candidates
        shape[0] = rng.choice([4, 8, 12])
        torch.manual_seed(trial)
        g = torch.randn(*shape)
        cands = [p for p in placement_pool if _valid(p[0], shape, ws)]
        src = rng.choice(cands)
        mid = rng.choice(cands)
        dst = rng.choice(cands)
        d = distribute_tensor(g, mesh, src)
        d = d.redistribute(placements=mid)
        d = d.redistribute(placements=dst)
        assert torch.allclose(d.full_tensor(), g, atol=1e-6), (
            trial, shape, src, mid, dst,
        )


def _valid(p, shape, ws):
    if isinstance(p, InterleavedShard):
        return shape[p.dim] % (p.interleaved_size * ws) == 0
    if isinstance(p, RaggedShard):
        if len(p.local_units) != ws:
            return False
        flat = 1
        for d in p.dims:
            flat *= shape[d]
        return flat % sum(p.local_units) == 0
    if isinstance(p, Shard):
        return p.dim < len(shape)
    return True


def test_redistribute_fuzz():
    spawn(2, _t_redistribute_fuzz)


def _t_uneven_from_local(rank, ws):
    import torch.distributed as dist

    mesh = init_device_mesh("cpu", (ws,))
    # rank 0 has 3 rows, rank 1 has 5 rows
    n = 3 if rank == 0 else 5
    local = torch.full((n, 4), float(rank))
    d = DTensor.from_local(local, mesh, [Shard(0)], support_uneven=True)
    assert d.shape == (8, 4)
    full = d.full_tensor()
    assert torch.equal(full[:3], torch.zeros(3, 4))
    assert torch.equal(full[3:], torch.ones(5, 4))


def test_uneven_from_local():
    spawn(2, _t_uneven_from_local)


def _t_explicit_collectives(rank, ws):
    from vescale_amd.dtensor import vescale_all_gather, vescale_all_reduce

    mesh = init_device_mesh("cpu", (ws,))
    g = torch.randn(8, 4)
    torch.manual_seed(3)
    g = torch.randn(8, 4)
    d = distribute_tensor(g, mesh, [Shard(0)])
    r = vescale_all_gather(d)
    assert all(p.is_replicate() for p in r.placements)
    assert torch.equal(r.to_local(), g)
    local = torch.full((2, 2), float(rank + 1))
    p = DTensor.from_local(local, mesh, [Partial()], shape=torch.Size((2, 2)))
    s = vescale_all_reduce(p)
    assert torch.equal(s.to_local(), torch.full((2, 2), 3.0))


def test_explicit_collectives():
    spawn(2, _t_explicit_collectives)


def test_rope_qkv_layout_cpu():
    """rope_qkv emits [B, H, S, D] (the attention kernels' native layout);
    CPU fallback must match an explicit slice+rope+permute reference."""
    import math

    import torch

    from vescale_amd.ops.functional import _rope_ref, build_rope_table, rope_qkv

    torch.manual_seed(3)
    B, S, Hq, Hkv, D = 2, 8, 4, 2, 16
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D)
    table = build_rope_table(S, D, 10000.0)
    q, k, v = rope_qkv(qkv, table, Hq, Hkv, D)
    assert q.shape == (B, Hq, S, D) and k.shape == (B, Hkv, S, D)
    qr = qkv[..., : Hq * D].reshape(B, S, Hq, D)
    kr = qkv[..., Hq * D : (Hq + Hkv) * D].reshape(B, S, Hkv, D)
    vr = qkv[..., (Hq + Hkv) * D :].reshape(B, S, Hkv, D)
    qe = _rope_ref(qr, table, 0, False).permute(0, 2, 1, 3)
    ke = _rope_ref(kr, table, 0, False).permute(0, 2, 1, 3)
    assert torch.allclose(q, qe, atol=1e-6)
    assert torch.allclose(k, ke, atol=1e-6)
    assert torch.allclose(v, vr.permute(0, 2, 1, 3), atol=1e-6)


def _t_conv_batch_parallel(rank, ws):
    """Batch-sharded conv2d: forward + backward parity vs single device
    (reference conv_ops.py behavior: Shard(0) input, Replicate weights,
    Partial weight grads)."""
    import torch

    from vescale_amd.dtensor import DTensor, init_device_mesh
    from vescale_amd.dtensor.placement_types import Replicate, Shard

    torch.manual_seed(7)
    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    conv = torch.nn.Conv2d(3, 8, 3, padding=1)
    x = torch.randn(4, 3, 8, 8, requires_grad=True)
    ref = conv(x)
    ref.sum().backward()
    ref_gw = conv.weight.grad.clone()
    ref_gx = x.grad.clone()
    conv.weight.grad = None
    conv.bias.grad = None

    xl = torch.chunk(x.detach(), ws)[rank].clone().requires_grad_()
    wl = conv.weight.detach().clone().requires_grad_()
    bl = conv.bias.detach().clone().requires_grad_()
    xd = DTensor.from_local(xl, mesh, [Shard(0)])
    wd = DTensor.from_local(wl, mesh, [Replicate()])
    bd = DTensor.from_local(bl, mesh, [Replicate()])
    out = torch.nn.functional.conv2d(xd, wd, bd, padding=1)
    assert isinstance(out._spec.placements[0], Shard)
    assert torch.allclose(out.full_tensor(), ref.detach(), atol=1e-5)
    out.sum().backward()
    # grads flow to the LOCAL leaves through _FromLocal (Partial weight
    # grads reduced to Replicate on the way out)
    assert torch.allclose(wl.grad, ref_gw, atol=1e-4)
    assert torch.allclose(xl.grad, torch.chunk(ref_gx, ws)[rank], atol=1e-5)


def test_conv_batch_parallel():
    from tests.common import spawn

    spawn(2, _t_conv_batch_parallel)


def _t_slice_select_backward(rank, ws):
    """slice_backward / select_backward on sharded DTensors (reference
    experimental_ops.py:27-76 coverage) flow grads to the local leaves."""
    import torch

    from vescale_amd.dtensor import DTensor, init_device_mesh
    from vescale_amd.dtensor.placement_types import Shard

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("D",))
    xl = torch.randn(4, 8, requires_grad=True)
    xd = DTensor.from_local(xl, mesh, [Shard(0)])
    xd[:, 2:6].sum().backward()
    want = torch.zeros(4, 8)
    want[:, 2:6] = 1
    assert torch.allclose(xl.grad, want)
    xl.grad = None
    xd.select(1, 3).sum().backward()
    want = torch.zeros(4, 8)
    want[:, 3] = 1
    assert torch.allclose(xl.grad, want)


def test_slice_select_backward():
    from tests.common import spawn

    spawn(2, _t_slice_select_backward)


def _t_amp_found_inf(rank, ws):
    """AMP grad-scaler inf scan on sharded DTensor grads: an inf on ONE
    rank's shard must set found_inf on ALL ranks (Partial("max") semantics,
    reference vescale/dtensor/_dispatch.py:60-117), and finite shards must
    be unscaled in place."""
    import torch.distributed as dist
    from vescale_amd.dtensor import Shard, distribute_tensor, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("DP",))
    g1 = distribute_tensor(torch.full((4, 8), 8.0), mesh, [Shard(0)])
    g2 = distribute_tensor(torch.full((6,), 2.0), mesh, [Shard(0)])
    if rank == 1:
        g1._local_tensor[0, 0] = float("inf")
    found_inf = torch.zeros(())
    inv_scale = torch.full((), 0.5)
    torch._amp_foreach_non_finite_check_and_unscale_(
        [g1, g2], found_inf, inv_scale
    )
    assert float(found_inf) == 1.0, f"rank {rank}: found_inf not reduced"
    assert torch.allclose(g2._local_tensor, torch.full_like(g2._local_tensor, 1.0))
    if rank == 0:
        assert torch.allclose(g1._local_tensor, torch.full_like(g1._local_tensor, 4.0))


def test_amp_found_inf_reduce():
    spawn(2, _t_amp_found_inf)


def _t_amp_gradscaler_flow(rank, ws):
    """Full torch.amp.GradScaler flow over a TP-sharded linear: scale ->
    backward -> unscale_ -> step skipped when an inf is injected on one
    rank, taken when grads are finite."""
    import torch.nn as nn
    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import Replicate, Shard, init_device_mesh

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("TP",))
    torch.manual_seed(0)
    net = nn.Linear(8, 8)
    plan = {
        "parameter": {r"weight": [Shard(0)], r"bias": [Shard(0)]},
        "forward": {"input": [[Replicate()]], "output": [[Replicate()]]},
    }
    parallelize_module(net, mesh, plan)
    opt = torch.optim.SGD(net.parameters(), lr=0.1)
    scaler = torch.amp.GradScaler("cpu", init_scale=4.0)

    # step 1: finite grads -> step taken
    w_before = net.weight._local_tensor.clone()
    loss = net(torch.randn(4, 8)).pow(2).mean()
    scaler.scale(loss).backward()
    scaler.unscale_(opt)
    for p in net.parameters():
        lg = p.grad._local_tensor if hasattr(p.grad, "_local_tensor") else p.grad
        assert torch.isfinite(lg).all()
    scaler.step(opt)
    scaler.update()
    opt.zero_grad()
    assert not torch.allclose(net.weight._local_tensor, w_before)

    # step 2: inf on rank 0's shard -> ALL ranks skip the step
    w_before = net.weight._local_tensor.clone()
    loss = net(torch.randn(4, 8)).pow(2).mean()
    scaler.scale(loss).backward()
    if rank == 0:
        net.weight.grad._local_tensor[0, 0] = float("inf")
    scaler.unscale_(opt)
    scaler.step(opt)
    scaler.update()
    assert torch.allclose(net.weight._local_tensor, w_before), (
        f"rank {rank} took a step on inf grads"
    )


def test_amp_gradscaler_end_to_end():
    spawn(2, _t_amp_gradscaler_flow)


# ---------------------------------------------------------------------------
# (_StridedRaggedShard, Shard) deep composition — 2D FSDPxTP as DTensor
# placements (reference placement_types.py:228 + docs/texts/raggedshard.md,
# re-specified clean: SRS composes AFTER the inner Shard, flattening the
# TP-local chunk; see api.distribute_tensor + the redistribute order_key)
# ---------------------------------------------------------------------------
def _t_srs_shard_compose(rank, ws):
    from vescale_amd.dtensor import _StridedRaggedShard

    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("dp", "tp"))
    coord = mesh.get_coordinate()
    dp, tp = coord[0], coord[1]
    w = torch.arange(48, dtype=torch.float32).reshape(8, 6)
    srs = _StridedRaggedShard(dims=(0,), local_units=(1, 1), split_factor=2)
    pl = [srs, Shard(0)]

    d = distribute_tensor(w, mesh, pl)
    # expected: TP chunk first (Shard(0) over 2 -> 4 rows), then flat halves
    tp_chunk = w[tp * 4 : (tp + 1) * 4]
    expect = tp_chunk.reshape(-1)[dp * 12 : (dp + 1) * 12]
    assert torch.equal(d._local_tensor, expect), (coord, d._local_tensor)

    # full gather back
    assert torch.equal(d.full_tensor(), w)

    # peel only the ragged layer: (SRS, S) -> (R, S) is the FSDP unshard
    unshard = d.redistribute(placements=[Replicate(), Shard(0)])
    assert torch.equal(unshard._local_tensor, tp_chunk)

    # and re-shard: (R, S) -> (SRS, S)
    reshard = unshard.redistribute(placements=pl)
    assert torch.equal(reshard._local_tensor, expect)

    # gradient path: (Partial, S) -> (SRS, S) reduces over dp then splits
    g = distribute_tensor(w, mesh, [Partial(), Shard(0)])
    gs = g.redistribute(placements=pl)
    assert torch.equal(gs._local_tensor, expect)

    # uneven units: dp0 holds 1 unit, dp1 holds 2 (of 3) — ragged proper
    srs_u = _StridedRaggedShard(dims=(0, 1), local_units=(1, 2), split_factor=2)
    d2 = distribute_tensor(w, mesh, [srs_u, Shard(0)])
    flat = tp_chunk.reshape(-1)
    expect2 = flat[:8] if dp == 0 else flat[8:]
    assert torch.equal(d2._local_tensor, expect2)
    assert torch.equal(d2.full_tensor(), w)


def test_srs_shard_compose():
    spawn(4, _t_srs_shard_compose)


def _t_to_local_grad_placements(rank, ws):
    """to_local(grad_placements=[Partial()]): the caller declares that the
    gradient flowing back is a PARTIAL sum (reference _api.py:410) — the
    wrapped grad then reduces on the way to the DTensor's layout."""
    mesh = init_device_mesh("cpu", (ws,))
    x = distribute_tensor(torch.ones(4), mesh, [Replicate()])
    x.requires_grad_(True)
    lt = x.to_local(grad_placements=[Partial()])
    # each rank contributes a DIFFERENT local grad; declared Partial means
    # the true grad is their SUM
    (lt * float(rank + 1)).sum().backward()
    g = x.grad
    gf = g.redistribute(placements=[Replicate()])._local_tensor
    expect = float(sum(r + 1 for r in range(ws)))
    assert torch.allclose(gf, torch.full((4,), expect)), gf


def test_to_local_grad_placements():
    spawn(2, _t_to_local_grad_placements)
