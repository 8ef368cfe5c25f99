"""Cross-mesh DTensor transfer (PP stage boundaries).

Parity: legacy/vescale/dtensor/redistribute.py:562 CrossMeshRedistribute +
the spec exchange at pipe/_schedules/pipedream_flush.py:71-115
(cross_mesh_send/recv): the sender ships (placements, global shape, dtype)
as an object message, then the local shard tensor p2p; the receiver
rebuilds the DTensor on ITS mesh.  Sender rank i of the source mesh pairs
with receiver rank i of the destination mesh (stage meshes are congruent
in the PP layout).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ._dtensor_spec import DTensorSpec
from .device_mesh import DeviceMesh
from .dtensor import DTensor
from .placement_types import Placement, Replicate, TensorMeta


def cross_mesh_send(dt: DTensor, dst_rank: int, tag: int = 0, group=None):
    meta = {
        "placements": tuple(dt._spec.placements),
        "shape": tuple(dt._spec.shape),
        "dtype": dt.dtype,
        "local_shape": tuple(dt._local_tensor.shape),
    }
    dist.send_object_list([meta], dst=dst_rank, group=group)
    dist.send(dt._local_tensor.contiguous(), dst=dst_rank, tag=tag, group=group)


def cross_mesh_recv(
    src_rank: int,
    dst_mesh: DeviceMesh,
    tag: int = 0,
    group=None,
    device: Optional[torch.device] = None,
) -> DTensor:
    holder = [None]
    dist.recv_object_list(holder, src=src_rank, group=group)
    meta = holder[0]
    local = torch.empty(
        meta["local_shape"], dtype=meta["dtype"],
        device=device or torch.device("cpu"),
    )
    dist.recv(local, src=src_rank, tag=tag, group=group)
    placements = meta["placements"]
    if len(placements) != dst_mesh.ndim:
        # placement arity differs across meshes: land as Replicate
        placements = tuple(Replicate() for _ in range(dst_mesh.ndim))
    st, acc = [], 1
    for s in reversed(meta["shape"]):
        st.append(acc)
        acc *= s
    tm = TensorMeta(torch.Size(meta["shape"]), tuple(reversed(st)), meta["dtype"])
    spec = DTensorSpec(dst_mesh, placements, tm)
    return DTensor(local, spec, requires_grad=False)
