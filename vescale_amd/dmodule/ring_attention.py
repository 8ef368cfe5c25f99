"""Ring attention — long-context SP with O(S/p) KV memory per GPU.

SURVEY §5.7 names ring attention as the second long-context extension
("a new schedule over the p2p layer"; the reference has neither it nor
Ulysses).  Layout matches ulysses.py: q/k/v arrive [B, H, S, D] as
DTensors Shard(seq) on a 1-D mesh.  Unlike Ulysses (full-sequence KV on
every rank after the a2a), ring attention keeps only S/p keys/values
resident: the KV block circulates the ring while each rank accumulates
its query block's attention with a numerically-stable running
log-sum-exp merge — mathematically exact, not an approximation.

xGMI fit: each ring step is ONE neighbor send+recv (batched isend/irecv
pair), so the p-step ring moves each KV byte over exactly one link per
hop — the per-link-bound traffic xGMI rings are sized for — and the
next block transfers while the current one computes (the sendrecv is
posted before the block's attention math).

Autograd: the ring shift is an autograd.Function whose backward shifts
gradients the OPPOSITE way around the ring; everything else is plain
differentiable torch, so grads are exact.  Every rank executes the same
graph, so the reverse-order backward sendrecvs pair up deadlock-free.
Memory note: this eager CPU/GPU reference keeps per-step activations
alive for autograd; a flash-style recompute backward is the kernel-level
follow-up (the HIP flash kernels in ops/csrc cover the dense core).

Causality is exact across blocks: with q-block index i and kv-block
index j (global offsets i*S_local, j*S_local), j>i contributes nothing
(skipped — the shift still runs), j==i uses the triangular mask, j<i is
unmasked.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from ..dtensor import DTensor, Shard

__all__ = ["ring_sdpa", "RingAttention"]


class _RingShift(torch.autograd.Function):
    """out_r = in_{(r-1) % p}: send my tensor to the next rank, receive
    from the previous.  Backward shifts grads the opposite direction."""

    @staticmethod
    def forward(ctx, t: torch.Tensor, group):
        ctx.group = group
        return _shift(t, group, forward=True)

    @staticmethod
    def backward(ctx, g: torch.Tensor):
        return _shift(g.contiguous(), ctx.group, forward=False), None


def _shift(t: torch.Tensor, group, *, forward: bool) -> torch.Tensor:
    ws = dist.get_world_size(group)
    me = dist.get_rank(group)
    nxt = dist.get_global_rank(group, (me + 1) % ws)
    prv = dist.get_global_rank(group, (me - 1) % ws)
    dst, src = (nxt, prv) if forward else (prv, nxt)
    t = t.contiguous()
    recv = torch.empty_like(t)  # empty_like of a contiguous t IS contiguous
    ops = [
        dist.P2POp(dist.isend, t, peer=dst, group=group),
        dist.P2POp(dist.irecv, recv, peer=src, group=group),
    ]
    for w in dist.batch_isend_irecv(ops):
        w.wait()
    return recv


def _merge(acc, lse, block_out, block_lse):
    """Running log-sum-exp merge of a new attention block (stable: both
    rescale factors are <= 1)."""
    new_lse = torch.logaddexp(lse, block_lse)
    acc = acc * torch.exp(lse - new_lse) + block_out * torch.exp(block_lse - new_lse)
    return acc, new_lse


def ring_sdpa(
    q: DTensor,
    k: DTensor,
    v: DTensor,
    *,
    is_causal: bool = False,
    scale: Optional[float] = None,
    seq_dim: int = 2,
) -> DTensor:
    """Exact sequence-parallel attention with ring-circulated KV.

    q/k/v: DTensors [B, H, S, D] Shard(seq_dim) on a 1-D mesh; returns
    the same layout.  GQA (fewer KV heads) works: KV heads broadcast over
    the query-head grouping like F.scaled_dot_product_attention with
    enable_gqa.
    """
    mesh = q.device_mesh
    assert mesh.ndim == 1, "ring_sdpa runs over a 1-D (SP) mesh"
    for name, t in (("q", q), ("k", k), ("v", v)):
        assert isinstance(t, DTensor) and t.placements[0].is_shard(seq_dim), (
            f"{name} must be Shard({seq_dim})"
        )
    group = mesh.get_group(0)
    p = dist.get_world_size(group)
    my = dist.get_rank(group)

    ql, kl, vl = q.to_local(), k.to_local(), v.to_local()
    B, Hq, Sq, D = ql.shape
    Hk = kl.shape[1]
    assert Hq % Hk == 0, "query heads must be a multiple of kv heads (GQA)"
    if Hk != Hq:
        rep = Hq // Hk
        kl = kl.repeat_interleave(rep, dim=1)
        vl = vl.repeat_interleave(rep, dim=1)
    sc = scale if scale is not None else 1.0 / math.sqrt(D)

    acc = torch.zeros_like(ql)
    lse = torch.full((B, Hq, Sq, 1), float("-inf"), dtype=ql.dtype, device=ql.device)
    # FINITE mask value, not -inf: every rank must compute EVERY block so
    # the autograd graph (incl. the ring-shift nodes) is identical across
    # ranks — a rank that skipped its last block would never run that
    # shift's backward and deadlock the ring.  exp(neg) underflows to 0,
    # so fully-masked blocks contribute nothing but still carry (zero)
    # gradient through k_cur/v_cur.
    neg = torch.finfo(ql.dtype).min / 2
    causal_mask = None
    if is_causal:
        causal_mask = torch.triu(
            torch.full((Sq, kl.shape[2]), neg, device=ql.device, dtype=ql.dtype),
            diagonal=1,
        )

    k_cur, v_cur = kl, vl
    for step in range(p):
        j = (my - step) % p  # global kv-block index currently held
        # post the next shift before the block math (on GPU the sendrecv
        # runs on the comm stream while the block computes)
        if step != p - 1:
            k_nxt = _RingShift.apply(k_cur, group)
            v_nxt = _RingShift.apply(v_cur, group)
        scores = (ql @ k_cur.transpose(-1, -2)) * sc
        if is_causal:
            if j == my:
                scores = scores + causal_mask
            elif j > my:
                scores = scores + neg  # future block: no contribution
        block_lse = scores.logsumexp(-1, keepdim=True)
        block_out = torch.softmax(scores, dim=-1) @ v_cur
        acc, lse = _merge(acc, lse, block_out, block_lse)
        if step != p - 1:
            k_cur, v_cur = k_nxt, v_nxt

    return DTensor.from_local(acc, mesh, [Shard(seq_dim)])


class RingAttention(torch.nn.Module):
    """Module wrapper mirroring UlyssesAttention."""

    def __init__(self, *, is_causal: bool = False, scale: Optional[float] = None):
        super().__init__()
        self.is_causal = is_causal
        self.scale = scale

    def forward(self, q: DTensor, k: DTensor, v: DTensor) -> DTensor:
        return ring_sdpa(q, k, v, is_causal=self.is_causal, scale=self.scale)
