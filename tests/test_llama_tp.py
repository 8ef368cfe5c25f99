"""Llama TP (Megatron f/g, plain tensors) + TPxFSDP 2D composition tests."""
import pytest
import torch

from tests.common import spawn

from vescale_amd.models.llama import LlamaModel, llama_tiny
from vescale_amd.models.llama_tp import shard_llama_state_dict


def _t_tp_parity(rank, ws):
    import torch.distributed as dist

    torch.manual_seed(5)
    cfg = llama_tiny()
    ref = LlamaModel(cfg)
    ref.init_weights()
    x = torch.randint(0, cfg.vocab_size, (2, 32))
    y = torch.roll(x, -1, dims=1)
    ref_loss = ref(x, y)
    ref_loss.backward()

    tp_model = LlamaModel(cfg, tp_group=dist.group.WORLD)
    full_sd = {k: v.detach() for k, v in ref.state_dict().items()}
    tp_model.load_state_dict(shard_llama_state_dict(full_sd, cfg, rank, ws))
    loss = tp_model(x, y)
    assert torch.allclose(loss, ref_loss.detach(), atol=2e-5), (float(loss), float(ref_loss))
    loss.backward()
    # grad parity: wqkv sharded (compare to ref shard), norm replicated
    gs = shard_llama_state_dict(
        {k: (v.grad if v.grad is not None else torch.zeros_like(v)) for k, v in ref.named_parameters()},
        cfg, rank, ws,
    )
    for name, p in tp_model.named_parameters():
        if p.grad is None:
            continue
        want = gs[name]
        assert torch.allclose(p.grad, want, atol=5e-5), (name, (p.grad - want).abs().max())


def test_llama_tp2_parity():
    spawn(2, _t_tp_parity)


def _t_tp_fsdp_2d(rank, ws):
    """TP=2 x FSDP=2 on 4 CPU ranks: loss parity vs single device."""
    import torch.distributed as dist

    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.fsdp import FSDP, FlatAdamW

    torch.manual_seed(5)
    cfg = llama_tiny()
    ref = LlamaModel(cfg)
    ref.init_weights()
    full_sd = {k: v.detach().clone() for k, v in ref.state_dict().items()}
    x = torch.randint(0, cfg.vocab_size, (4, 32))
    y = torch.roll(x, -1, dims=1)

    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("DP", "TP"))
    coord = mesh.get_coordinate()
    dp_rank, tp_rank = coord
    tp_group = mesh.get_group(1)
    dp_group = mesh.get_group(0)

    model = LlamaModel(cfg, tp_group=tp_group)
    model.load_state_dict(shard_llama_state_dict(full_sd, cfg, tp_rank, 2))
    eng = FSDP(model, mesh, mesh_dim=0, param_dtype=torch.float32,
               device=torch.device("cpu"))
    opt = FlatAdamW(eng, lr=1e-3, weight_decay=0.0)

    # reference: full-batch loss/step on single device
    ref_opt_losses = []
    ropt = torch.optim.AdamW(ref.parameters(), lr=1e-3, betas=(0.9, 0.95),
                             eps=1e-8, weight_decay=0.0)
    for step in range(3):
        ropt.zero_grad()
        l = ref(x, y)
        l.backward()
        ropt.step()
        ref_opt_losses.append(float(l))

    losses = []
    for step in range(3):
        xs = torch.chunk(x, 2)[dp_rank]
        ys = torch.chunk(y, 2)[dp_rank]
        loss = eng(xs, ys)
        loss.backward()
        opt.step()
        g = loss.detach().clone()
        dist.all_reduce(g, group=dp_group)
        losses.append(float(g) / 2)
    for a, b in zip(losses, ref_opt_losses):
        assert abs(a - b) < 3e-3, (losses, ref_opt_losses)


def test_llama_tp2_fsdp2():
    spawn(4, _t_tp_fsdp_2d)
