"""Ulysses-style sequence-parallel attention (long-context SP).

SURVEY §5.7: the reference scales sequence length only via Megatron-SP
activation sharding; it has no Ulysses/ring attention.  On MI355X the
idiomatic long-context extension is exactly the DTensor redistribute the
planner already owns: Shard(seq) <-> Shard(head) is a SINGLE all_to_all
per tensor (see redistribute.py S(a)->S(b); pinned by
tests/test_edges.py::test_shard_to_shard_single_a2a), so Ulysses is four
uneven-free all-to-alls around a fully-local attention:

    [B, H, S/p, D]  --a2a-->  [B, H/p, S, D]   (q, k, v)
    local attention over the FULL sequence     (exact causality)
    [B, H/p, S, D]  --a2a-->  [B, H, S/p, D]   (out)

On an 8-GPU xGMI node the a2a rides all 7 links concurrently per GPU
(fully-connected point-to-point), which is precisely the traffic shape
xGMI is best at — no ring serialization.

Autograd flows through redistribute/from_local, so the same four
all-to-alls run (transposed) in backward.

Requires H % p == 0 and S % p == 0 on the sharded mesh dim.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ..dtensor import DTensor, Shard

__all__ = ["ulysses_sdpa", "UlyssesAttention"]


def ulysses_sdpa(
    q: DTensor,
    k: DTensor,
    v: DTensor,
    *,
    is_causal: bool = False,
    scale: Optional[float] = None,
    seq_dim: int = 2,
    head_dim: int = 1,
) -> DTensor:
    """Sequence-parallel scaled-dot-product attention.

    q/k/v: DTensors of layout [B, H, S, D] (default dims) sharded
    Shard(seq_dim) on a 1-D mesh.  Returns a DTensor sharded the same
    way.  Head-count may differ between q and k/v (GQA) as long as both
    divide the mesh size.
    """
    mesh = q.device_mesh
    assert mesh.ndim == 1, "ulysses_sdpa runs over a 1-D (SP) mesh"
    for name, t in (("q", q), ("k", k), ("v", v)):
        assert isinstance(t, DTensor), f"{name} must be a DTensor"
        p = t.placements[0]
        assert p.is_shard(seq_dim), (
            f"{name} must be Shard({seq_dim}) (sequence-sharded), got {p}"
        )

    # seq -> head: one all_to_all per tensor
    qh = q.redistribute(placements=[Shard(head_dim)])
    kh = k.redistribute(placements=[Shard(head_dim)])
    vh = v.redistribute(placements=[Shard(head_dim)])

    ql, kl, vl = qh.to_local(), kh.to_local(), vh.to_local()
    o = F.scaled_dot_product_attention(
        ql, kl, vl,
        is_causal=is_causal, scale=scale,
        enable_gqa=(ql.shape[1] != kl.shape[1]),
    )
    oh = DTensor.from_local(o, mesh, [Shard(head_dim)])
    # head -> seq: the fourth all_to_all
    return oh.redistribute(placements=[Shard(seq_dim)])


class UlyssesAttention(torch.nn.Module):
    """Module wrapper: drop-in for an SDPA core inside a sequence-parallel
    block (activations arrive Shard(seq), leave Shard(seq))."""

    def __init__(self, *, is_causal: bool = False, scale: Optional[float] = None):
        super().__init__()
        self.is_causal = is_causal
        self.scale = scale

    def forward(self, q: DTensor, k: DTensor, v: DTensor) -> DTensor:
        return ulysses_sdpa(q, k, v, is_causal=self.is_causal, scale=self.scale)
