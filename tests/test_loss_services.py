"""loss_parallel, vocab-parallel patches, CommDebugMode, VeDeviceMesh."""
import pytest
import torch
import torch.nn.functional as F

from tests.common import spawn

from vescale_amd.dtensor import (
    DTensor,
    Replicate,
    Shard,
    distribute_tensor,
    init_device_mesh,
)


def _t_loss_parallel(rank, ws):
    from vescale_amd.dtensor.loss import loss_parallel

    torch.manual_seed(31)
    logits = torch.randn(8, 12)
    tgt = torch.randint(0, 12, (8,))
    tgt[0] = -100
    ref = F.cross_entropy(logits, tgt, ignore_index=-100)
    logits_r = logits.clone().requires_grad_(True)
    F.cross_entropy(logits_r, tgt, ignore_index=-100).backward()

    mesh = init_device_mesh("cpu", (ws,))
    d = distribute_tensor(logits, mesh, [Shard(1)]).requires_grad_(True)
    with loss_parallel():
        loss = F.cross_entropy(d, tgt, ignore_index=-100)
        lv = loss.to_local() if isinstance(loss, DTensor) else loss
        assert torch.allclose(lv, ref, atol=1e-5), (float(lv), float(ref))
        loss.backward()
    g = d.grad.full_tensor()
    assert torch.allclose(g, logits_r.grad, atol=1e-5)


def test_loss_parallel():
    spawn(2, _t_loss_parallel)


def _t_vp_cross_entropy(rank, ws):
    from vescale_amd.model import VocabParallelCrossEntropy

    torch.manual_seed(37)
    mesh = init_device_mesh("cpu", (ws,))
    logits = torch.randn(6, 16)
    tgt = torch.randint(0, 16, (6,))
    ref = F.cross_entropy(logits, tgt, reduction="none")
    d = distribute_tensor(logits, mesh, [Shard(1)])
    ce = VocabParallelCrossEntropy(mesh)
    loss = ce(d, tgt)
    assert torch.allclose(loss.float(), ref, atol=1e-5)


def test_vp_cross_entropy():
    spawn(2, _t_vp_cross_entropy)


def _t_vp_embedding(rank, ws):
    from vescale_amd.model import VocabParallelEmbedding

    torch.manual_seed(41)
    mesh = init_device_mesh("cpu", (ws,))
    emb = torch.nn.Embedding(20, 8)
    idx = torch.randint(0, 20, (3, 5))
    ref = emb(idx)
    vp = VocabParallelEmbedding(emb, mesh)
    out = vp(idx)
    out_l = out.to_local() if isinstance(out, DTensor) else out
    assert torch.allclose(out_l, ref.detach(), atol=1e-5)


def test_vp_embedding():
    spawn(2, _t_vp_embedding)


def _t_comm_mode(rank, ws):
    from vescale_amd.debug import CommDebugMode

    mesh = init_device_mesh("cpu", (ws,))
    g = torch.randn(8, 4)
    d = distribute_tensor(g, mesh, [Shard(0)])
    with CommDebugMode() as cm:
        d.redistribute(placements=[Replicate()])
    assert cm.total >= 1
    assert cm.get_comm_counts().get("mesh_all_gather", 0) >= 1


def test_comm_mode():
    spawn(2, _t_comm_mode)


def _t_vedevicemesh(rank, ws):
    from vescale_amd.devicemesh_api import VESCALE_DEVICE_MESH as V

    V.init_device_mesh("cpu", (2, 2), mesh_dim_names=("DP", "TP"))
    assert V.get_data_parallel_world_size() == 2
    assert V.get_tensor_parallel_world_size() == 2
    assert V.get_pipeline_parallel_world_size() == 1
    coord_dp = V.get_data_parallel_rank()
    coord_tp = V.get_tensor_parallel_rank()
    assert rank == coord_dp * 2 + coord_tp
    assert V.is_first_stage() and V.is_last_stage()
    tp_mesh = V.get_tensor_parallel_mesh()
    assert tp_mesh.size() == 2


def test_vedevicemesh():
    spawn(4, _t_vedevicemesh)
