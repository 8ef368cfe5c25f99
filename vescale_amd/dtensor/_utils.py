"""DTensor utilities (parity: legacy/vescale/dtensor/_utils.py:51-463).

compute_local_shape_and_global_offset / gather_local_tensor_shape /
compute_global_tensor_info / equal / allclose.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ._dtensor_spec import DTensorSpec
from .device_mesh import DeviceMesh
from .placement_types import InterleavedShard, Partial, Placement, RaggedShard, Replicate, Shard


def compute_local_shape_and_global_offset(
    global_shape: Sequence[int],
    mesh: DeviceMesh,
    placements: Sequence[Placement],
    coord: Optional[Sequence[int]] = None,
) -> Tuple[Tuple[int, ...], Tuple[int, ...]]:
    """(local shape, global offsets) of this rank's shard (reference :51)."""
    if coord is None:
        coord = mesh.get_coordinate()
    assert coord is not None
    shape = list(global_shape)
    offsets = [0] * len(shape)
    for md, p in enumerate(placements):
        w = mesh.size(md)
        idx = coord[md]
        if isinstance(p, RaggedShard):
            n = p.local_numel(tuple(shape), idx)
            off = p.local_offset_numel(tuple(shape), idx)
            return (n,), (off,)
        if isinstance(p, InterleavedShard):
            shape = list(p.local_shape(shape, w, idx))
        elif isinstance(p, Shard):
            total = shape[p.dim]
            offsets[p.dim] += Shard.chunk_offset(total, w, idx)
            shape[p.dim] = Shard.chunk_size(total, w, idx)
    return tuple(shape), tuple(offsets)


def gather_local_tensor_shape(
    local: torch.Tensor, mesh: DeviceMesh, mesh_dim: int = 0
) -> List[Tuple[int, ...]]:
    """All ranks' local shapes on a mesh dim (uneven-shard metadata
    exchange; reference :133)."""
    pg = mesh.get_group(mesh_dim)
    shapes: List = [None] * mesh.size(mesh_dim)
    dist.all_gather_object(shapes, tuple(local.shape), group=pg)
    return shapes


def compute_global_tensor_info(
    local: torch.Tensor, mesh: DeviceMesh, placements: Sequence[Placement]
) -> Tuple[List[int], List[int]]:
    """Infer global (shape, stride) from a local tensor + placements
    (reference :168, even-shard assumption)."""
    shape = list(local.shape)
    for md, p in enumerate(placements):
        w = mesh.size(md)
        if isinstance(p, (Shard, InterleavedShard)):
            shape[p.dim] *= w
        elif isinstance(p, RaggedShard):
            raise ValueError("cannot infer global shape from a ragged local")
    stride, acc = [], 1
    for s in reversed(shape):
        stride.append(acc)
        acc *= s
    return shape, list(reversed(stride))


def equal(a, b) -> bool:
    """Global equality of two DTensors (reference :374)."""
    from .dtensor import DTensor

    if isinstance(a, DTensor) and isinstance(b, DTensor):
        return torch.equal(a, b)  # routed via the aten.equal bypass
    return torch.equal(a, b)


def allclose(a, b, rtol: float = 1e-5, atol: float = 1e-8) -> bool:
    """Global allclose (reference :388): compares full tensors."""
    from .dtensor import DTensor

    if isinstance(a, DTensor):
        a = a.full_tensor()
    if isinstance(b, DTensor):
        b = b.full_tensor()
    return torch.allclose(a, b, rtol=rtol, atol=atol)
