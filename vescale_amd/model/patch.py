"""Vocab-parallel module patches.

Parity: legacy/vescale/model/patch/vp_embedding.py:38 (range mask +
partial + allreduce) and vp_cross_entropy.py:43-149 (local max ->
allreduce-max -> masked gather -> allreduce-sum log-sum-exp), as opt-in
module-level rewrites for models whose embedding/classifier weights are
vocab-sharded Shard(0).
"""
from __future__ import annotations


import torch
import torch.distributed as dist
import torch.nn as nn

from ..dtensor import DeviceMesh, DTensor, Partial, Replicate, Shard
from ..dtensor import _collective_utils as cc
from ..dtensor.placement_types import TensorMeta
from ..dtensor._dtensor_spec import DTensorSpec


class VocabParallelEmbedding(nn.Module):
    """Embedding with Shard(0) (vocab-sharded) weight; output allreduced.

    Accepts a plain nn.Embedding to wrap; the weight may already be a
    DTensor (from parallelize_module) or a plain tensor to distribute.
    """

    def __init__(self, emb: nn.Embedding, mesh: DeviceMesh, mesh_dim: int = 0):
        super().__init__()
        self.mesh = mesh
        self.mesh_dim = mesh_dim
        self.num_embeddings = emb.num_embeddings
        w = emb.weight
        if not isinstance(w.data, DTensor):
            from ..dtensor import distribute_tensor

            d = distribute_tensor(w.data, mesh, [Shard(0)] + [Replicate()] * (mesh.ndim - 1))
            self.weight = nn.Parameter(d, requires_grad=w.requires_grad)
        else:
            self.weight = w

    def forward(self, idx: torch.Tensor) -> torch.Tensor:
        import torch.nn.functional as F

        if not isinstance(idx, DTensor):
            idx = DTensor.from_local(idx, self.mesh, [Replicate()] * self.mesh.ndim)
        out = F.embedding(idx, self.weight)  # vocab-parallel handler -> Partial
        if isinstance(out, DTensor) and any(p.is_partial() for p in out.placements):
            out = out.redistribute(placements=[
                Replicate() if p.is_partial() else p for p in out.placements
            ])
        return out


class _VocabParallelCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits_local, target, start, vocab_total, mesh, mesh_dim):
        # logits_local: [N, V/w] raw logits shard
        mx = logits_local.amax(dim=-1, keepdim=True)
        cc.mesh_all_reduce(mx, mesh, "max", mesh_dim)
        x = (logits_local - mx).float()
        ex = x.exp()
        s = ex.sum(dim=-1, keepdim=True)
        cc.mesh_all_reduce(s, mesh, "sum", mesh_dim)
        logz = s.log()
        n = logits_local.shape[-1]
        inrange = (target >= start) & (target < start + n)
        shifted = (target - start).clamp(0, max(0, n - 1))
        picked = x.gather(-1, shifted.unsqueeze(-1)).squeeze(-1)
        picked = torch.where(inrange, picked, torch.zeros_like(picked))
        t = picked.clone()
        cc.mesh_all_reduce(t, mesh, "sum", mesh_dim)
        loss = (logz.squeeze(-1) - t)
        ctx.save_for_backward(ex, s, shifted, inrange)
        ctx.meta = (mesh, mesh_dim)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        ex, s, shifted, inrange = ctx.saved_tensors
        softmax = ex / s
        g = grad_out.unsqueeze(-1)
        dx = softmax * g
        sub = torch.where(
            inrange, g.squeeze(-1), torch.zeros_like(g.squeeze(-1))
        )
        dx.scatter_add_(-1, shifted.unsqueeze(-1), -sub.unsqueeze(-1))
        return dx.to(ex.dtype), None, None, None, None, None


class VocabParallelCrossEntropy(nn.Module):
    """CE over vocab-sharded raw logits — per-token loss vector output
    (reduction applied by caller), numerically matching dense CE."""

    def __init__(self, mesh: DeviceMesh, mesh_dim: int = 0):
        super().__init__()
        self.mesh = mesh
        self.mesh_dim = mesh_dim

    def forward(self, logits, target):
        if isinstance(logits, DTensor):
            spec = logits._spec
            vocab = spec.shape[-1]
            md = None
            for i, p in enumerate(spec.placements):
                if isinstance(p, Shard) and p.dim == spec.ndim - 1:
                    md = i
            assert md is not None, "logits must be vocab-sharded"
            w = self.mesh.size(md)
            r = self.mesh.get_local_rank(md)
            start = Shard.chunk_offset(vocab, w, r)
            local = logits._local_tensor
            loss = _VocabParallelCE.apply(local, target if not isinstance(target, DTensor) else target.to_local(), start, vocab, self.mesh, md)
            return loss
        # dense fallback
        import torch.nn.functional as F

        return F.cross_entropy(logits, target, reduction="none")
