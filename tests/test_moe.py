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
