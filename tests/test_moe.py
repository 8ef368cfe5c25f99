"""MoE/EP tests: dense vs expert-parallel parity (mirrors
legacy/test/parallel/ddp_optim/test_moe.py + mixtral_EP_training)."""
import pytest
import torch

from tests.common import spawn

from vescale_amd.models.mixtral import MixtralModel, MoELayer, mixtral_tiny


def _t_ep_forward_parity(rank, ws):
    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.moe import parallelize_experts

    torch.manual_seed(9)
    cfg = mixtral_tiny()
    dense = MixtralModel(cfg)
    dense.init_weights()
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    y = torch.roll(x, -1, dims=1)
    ref_loss = dense(x, y)
    ref_loss.backward()

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("EP",))
    torch.manual_seed(9)
    model = MixtralModel(cfg)
    model.init_weights()
    parallelize_experts(model, mesh)
    loss = model(x, y)
    assert torch.allclose(loss, ref_loss.detach(), atol=1e-5), (float(loss), float(ref_loss))
    loss.backward()
    # expert grads match the dense run for locally-owned experts
    for li, layer in enumerate(model.layers):
        moe = layer.moe
        for e in moe.local_expert_ids:
            g_ep = moe.experts[e].w13.weight.grad
            g_ref = dense.layers[li].moe.experts[e].w13.weight.grad
            if g_ref is None:
                assert g_ep is None or g_ep.abs().max() == 0
                continue
            assert g_ep is not None, (li, e)
            # each EP rank fed the same batch, so the expert saw ws x the
            # tokens: grad = ws * dense grad (DPxEP accumulation semantics)
            assert torch.allclose(g_ep, ws * g_ref, atol=1e-5), (
                li, e, (g_ep - ws * g_ref).abs().max(),
            )
    # router grads are dense on every rank
    g_r = model.layers[0].moe.router.weight.grad
    g_rr = dense.layers[0].moe.router.weight.grad
    assert torch.allclose(g_r, g_rr, atol=1e-5)


def test_ep_parity_ws2():
    spawn(2, _t_ep_forward_parity)


def test_allocator_dispatcher():
    from vescale_amd.moe import BasicExpertsAllocator, BasicTokenDispatcher

    alloc = BasicExpertsAllocator(8, 4)
    assert [alloc.owner_of(e) for e in range(8)] == [0, 0, 1, 1, 2, 2, 3, 3]
    assert alloc.experts_of(1) == [2, 3]
    disp = BasicTokenDispatcher(alloc)
    ids = torch.tensor([7, 0, 3, 3, 5, 1])
    perm, splits, sorted_e = disp.dispatch_plan(ids)
    assert sum(splits) == 6
    assert splits == [2, 2, 1, 1]
    assert sorted_e.tolist() == [0, 1, 3, 3, 5, 7]


def test_global_all_to_all_grad():
    from vescale_amd.moe import global_all_to_all_single

    x = torch.randn(6, 4, requires_grad=True)
    out = global_all_to_all_single(x, None, None, None)
    out.sum().backward()
    assert torch.allclose(x.grad, torch.ones_like(x))


def _t_dynamic_rebalance(rank, ws):
    """Dynamic expert re-placement: params + optimizer state MOVE between
    ranks and the EP forward still matches the dense reference after."""
    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.moe import (
        LoadBalancedExpertsAllocator,
        parallelize_experts,
        rebalance_experts,
    )

    torch.manual_seed(11)
    cfg = mixtral_tiny()
    dense = MixtralModel(cfg)
    dense.init_weights()
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    y = torch.roll(x, -1, dims=1)

    mesh = init_device_mesh("cpu", (ws,), mesh_dim_names=("EP",))
    torch.manual_seed(11)
    model = MixtralModel(cfg)
    model.init_weights()
    parallelize_experts(model, mesh, allocator_cls=LoadBalancedExpertsAllocator)

    moe = model.layers[0].moe
    opt = torch.optim.Adam(
        [p for p in moe.parameters() if p.requires_grad], lr=1e-3
    )
    # one step to create optimizer state for the local experts
    loss = model(x, y)
    loss.backward()
    opt.step()
    opt.zero_grad()
    dense_loss = dense(x, y)
    dense_loss.backward()
    dense_opt = torch.optim.Adam(dense.layers[0].moe.parameters(), lr=1e-3)
    dense_opt.step()
    dense_opt.zero_grad()

    E = cfg.n_experts
    old_owner = [moe.allocator.owner_of(e) for e in range(E)]
    # snapshot every expert's post-step weights on all ranks (broadcast
    # from the owner) so moved params can be checked BITWISE after
    import torch.distributed as dist
    template = moe.make_expert()
    pnames = [n for n, _ in template.named_parameters()]
    snap = {}
    for e in range(E):
        src = old_owner[e]
        snap[e] = {}
        for n in pnames:
            if rank == src:
                t = dict(moe.experts[e].named_parameters())[n].detach().clone()
            else:
                t = torch.empty(dict(template.named_parameters())[n].shape)
            dist.broadcast(t, src=src)
            snap[e][n] = t
    # skewed load: the two hottest experts both start on rank 0 (blocked
    # placement), so LPT must split them across ranks -> a move
    counts = [float(e) for e in range(E)]
    counts[0] = 1000.0
    counts[1] = 900.0
    changed = rebalance_experts(moe, counts, optimizer=opt)
    assert changed, "placement should have changed under skewed load"
    new_owner = [moe.allocator.owner_of(e) for e in range(E)]
    assert new_owner != old_owner
    moved = [e for e in range(E) if new_owner[e] != old_owner[e]]
    assert moved

    # moved experts: params on the new owner are BITWISE the source's
    # post-step values (transfer fidelity), and optimizer state arrived
    for e in moved:
        if rank == new_owner[e]:
            for n, p in moe.experts[e].named_parameters():
                assert torch.equal(p.detach(), snap[e][n]), (e, n)
            got = moe.experts[e].w13.weight.detach()
            st = opt.state.get(moe.experts[e].w13.weight, {})
            dst = dense_opt.state[dense.layers[0].moe.experts[e].w13.weight]
            if "exp_avg" in dst:
                assert "exp_avg" in st, e
                # EP grads were ws x dense (same batch on every rank);
                # loose tol: only the SHAPE/arrival and scale matter here
                assert torch.allclose(st["exp_avg"], ws * dst["exp_avg"], atol=1e-5)
        if rank == old_owner[e]:
            assert type(moe.experts[e]).__name__ == "_RemoteExpert"

    # local ids consistent with allocator
    assert moe.local_expert_ids == moe.allocator.experts_of(rank)

    # EP forward after re-placement matches a dense layer rebuilt from
    # the snapshot weights (bitwise-same expert params, same router)
    ref = dense.layers[0].moe
    with torch.no_grad():
        for e in range(E):
            dref = dict(ref.experts[e].named_parameters())
            for n in pnames:
                dref[n].copy_(snap[e][n])
        ref.router.weight.copy_(moe.router.weight)
        torch.manual_seed(123)
        h = torch.randn(2, 16, cfg.dim)
        out_ep = moe(h)
        out_ref = ref(h)
    assert torch.allclose(out_ep, out_ref, atol=1e-5), (
        (out_ep - out_ref).abs().max()
    )


def test_dynamic_expert_rebalance():
    spawn(2, _t_dynamic_rebalance)
