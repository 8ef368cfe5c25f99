"""Public DTensor API: distribute_tensor, from_local, to_local,
redistribute, explicit collective helpers.

Parity: legacy/vescale/dtensor/api.py:39-435 + vescale/dtensor/_api.py:589.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.distributed as dist

from . import _collective_utils as cc
from ._dtensor_spec import DTensorSpec, make_spec
from .device_mesh import DeviceMesh
from .dtensor import DTensor
from .placement_types import (
    InterleavedShard,
    _StridedRaggedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
)

__all__ = [
    "distribute_tensor",
    "from_local",
    "to_local",
    "redistribute_dtensor",
    "normalize_placements",
]


def normalize_placements(placements, mesh_ndim: int):
    if placements is None:
        return tuple(Replicate() for _ in range(mesh_ndim))
    out = tuple(placements)
    assert len(out) == mesh_ndim
    return out


def distribute_tensor(
    tensor: torch.Tensor,
    device_mesh: DeviceMesh,
    placements: Optional[Sequence[Placement]] = None,
) -> DTensor:
    """Distribute a full (global) tensor present on every rank (or on rank 0
    for scatter-based placements) into a DTensor.

    All ranks must call this with the same global tensor value (standard
    SPMD contract); sharded placements take the local chunk with NO
    communication; RaggedShard takes the local flat range.
    """
    placements = normalize_placements(placements, device_mesh.ndim)
    coord = device_mesh.get_coordinate()
    local = tensor
    if coord is not None:
        # _StridedRaggedShard composes AFTER every other mesh dim: it ragged-
        # flattens the already-sharded (e.g. TP-local) chunk, the 2D FSDPxTP
        # composition (reference placement_types.py:228, re-specified clean:
        # docs in placement_types._StridedRaggedShard).
        mds = sorted(
            range(len(placements)),
            key=lambda i: (isinstance(placements[i], _StridedRaggedShard), i),
        )
        for md in mds:
            p = placements[md]
            w = device_mesh.size(md)
            if isinstance(p, RaggedShard):
                local = p.split_tensor(local, w)[coord[md]]
            elif isinstance(p, InterleavedShard):
                local = p.split_tensor(local, w)[coord[md]]
            elif isinstance(p, Shard):
                off = Shard.chunk_offset(local.size(p.dim), w, coord[md])
                sz = Shard.chunk_size(local.size(p.dim), w, coord[md])
                local = local.narrow(p.dim, off, sz).contiguous()
            elif isinstance(p, Partial):
                if coord[md] != 0:
                    local = torch.zeros_like(local)
    st, acc = [], 1
    for s in reversed(list(tensor.shape)):
        st.append(acc)
        acc *= s
    tm = TensorMeta(tensor.shape, tuple(reversed(st)), tensor.dtype)
    spec = DTensorSpec(device_mesh, placements, tm)
    d = DTensor(local.detach().clone() if local is tensor else local, spec,
                requires_grad=tensor.requires_grad)
    return d


def from_local(*args, **kwargs) -> DTensor:
    return DTensor.from_local(*args, **kwargs)


def to_local(dt: DTensor) -> torch.Tensor:
    return dt.to_local()


def redistribute_dtensor(
    dt: DTensor, device_mesh: Optional[DeviceMesh] = None, placements=None, async_op: bool = False
) -> DTensor:
    return dt.redistribute(device_mesh, placements, async_op=async_op)


# explicit collective APIs (parity: legacy api.py:314-433)
def vescale_all_gather(dt: DTensor, mesh_dims=None, async_op: bool = False) -> DTensor:
    spec = dt._spec
    mesh = spec.mesh
    dims = range(mesh.ndim) if mesh_dims is None else (
        [mesh_dims] if isinstance(mesh_dims, int) else mesh_dims
    )
    placements = list(spec.placements)
    for md in dims:
        if not placements[md].is_replicate() and not placements[md].is_partial():
            placements[md] = Replicate()
    return dt.redistribute(placements=placements, async_op=async_op)


def vescale_all_reduce(dt: DTensor, mesh_dims=None, async_op: bool = False) -> DTensor:
    spec = dt._spec
    mesh = spec.mesh
    dims = range(mesh.ndim) if mesh_dims is None else (
        [mesh_dims] if isinstance(mesh_dims, int) else mesh_dims
    )
    placements = list(spec.placements)
    for md in dims:
        if placements[md].is_partial():
            placements[md] = Replicate()
    return dt.redistribute(placements=placements, async_op=async_op)


def vescale_reduce_scatter(
    dt: DTensor, scatter_dim: int, mesh_dims=None, async_op: bool = False
) -> DTensor:
    spec = dt._spec
    mesh = spec.mesh
    dims = range(mesh.ndim) if mesh_dims is None else (
        [mesh_dims] if isinstance(mesh_dims, int) else mesh_dims
    )
    placements = list(spec.placements)
    for md in dims:
        if placements[md].is_partial():
            placements[md] = Shard(scatter_dim)
    return dt.redistribute(placements=placements, async_op=async_op)
