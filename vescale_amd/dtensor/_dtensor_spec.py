"""DTensorSpec: (mesh, placements, tensor_meta) — the complete description
of how a global tensor is laid out across a DeviceMesh.

Parity: legacy/vescale/dtensor/placement_types.py (DTensorSpec) and
vescale/dtensor/_dtensor_spec.py.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import torch

from .device_mesh import DeviceMesh
from .placement_types import (
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
)


@dataclass
class DTensorSpec:
    mesh: DeviceMesh
    placements: Tuple[Placement, ...]
    tensor_meta: Optional[TensorMeta] = None

    def __post_init__(self):
        self.placements = tuple(self.placements)
        assert len(self.placements) == self.mesh.ndim, (
            f"{len(self.placements)} placements for {self.mesh.ndim}-d mesh"
        )
        self._hash = None

    # specs are hashed as sharding-prop cache keys
    def _hash_key(self):
        tm = self.tensor_meta
        return (
            self.mesh,
            self.placements,
            (tuple(tm.shape), tuple(tm.stride), tm.dtype) if tm else None,
        )

    def __hash__(self):
        if self._hash is None:
            self._hash = hash(self._hash_key())
        return self._hash

    def __eq__(self, other):
        return (
            isinstance(other, DTensorSpec)
            and self.mesh == other.mesh
            and self.placements == other.placements
            and self.tensor_meta == other.tensor_meta
        )

    def __repr__(self):
        return f"Spec({list(self.placements)}, shape={tuple(self.tensor_meta.shape) if self.tensor_meta else None})"

    # ------------------------------------------------------------------
    @property
    def shape(self) -> torch.Size:
        assert self.tensor_meta is not None
        return self.tensor_meta.shape

    @property
    def ndim(self) -> int:
        return len(self.shape)

    @property
    def dtype(self) -> torch.dtype:
        assert self.tensor_meta is not None
        return self.tensor_meta.dtype

    def bytes(self) -> int:
        tm = self.tensor_meta
        if tm is None:
            return 0
        n = 1
        for s in tm.shape:
            n *= s
        return n * tm.dtype.itemsize

    @property
    def is_replicated(self) -> bool:
        return all(p.is_replicate() for p in self.placements)

    @property
    def is_sharded(self) -> bool:
        return any(p.is_shard() or p.is_interleaved_shard() or p.is_ragged_shard() for p in self.placements)

    @property
    def is_partial(self) -> bool:
        return any(p.is_partial() for p in self.placements)

    def has_ragged(self) -> bool:
        return any(p.is_ragged_shard() for p in self.placements)

    def dim_map(self) -> List[int]:
        """tensor dim -> mesh dim sharding map (-1 = not sharded).  Ragged
        placements map their first flattened dim."""
        m = [-1] * self.ndim
        for mesh_dim, p in enumerate(self.placements):
            if isinstance(p, (Shard, InterleavedShard)):
                assert m[p.dim] == -1, f"dim {p.dim} sharded on two mesh dims"
                m[p.dim] = mesh_dim
            elif isinstance(p, RaggedShard):
                m[p.dims[0]] = mesh_dim
        return m

    def num_shards_on_dim(self, tensor_dim: int) -> int:
        n = 1
        for mesh_dim, p in enumerate(self.placements):
            if isinstance(p, (Shard, InterleavedShard)) and p.dim == tensor_dim:
                n *= self.mesh.size(mesh_dim)
        return n

    # ------------------------------------------------------------------
    def local_shape(self, coord: Optional[Sequence[int]] = None) -> Tuple[int, ...]:
        """Shape of this rank's local tensor.  For RaggedShard, the local
        tensor is flat 1-D."""
        assert self.tensor_meta is not None
        if coord is None:
            coord = self.mesh.get_coordinate()
        assert coord is not None, "rank not in mesh"
        shape = list(self.tensor_meta.shape)
        for mesh_dim, p in enumerate(self.placements):
            idx = coord[mesh_dim]
            w = self.mesh.size(mesh_dim)
            if isinstance(p, RaggedShard):
                return (p.local_numel(tuple(shape), idx),)
            if isinstance(p, InterleavedShard):
                shape = list(p.local_shape(shape, w, idx))
            elif isinstance(p, Shard):
                shape = list(p.local_shape(shape, w, idx))
        return tuple(shape)

    def local_offsets(self, coord: Optional[Sequence[int]] = None) -> Tuple[int, ...]:
        """Global offsets of this rank's shard (Shard placements only;
        reference _utils.py:51 compute_local_shape_and_global_offset)."""
        assert self.tensor_meta is not None
        if coord is None:
            coord = self.mesh.get_coordinate()
        assert coord is not None
        shape = list(self.tensor_meta.shape)
        offsets = [0] * len(shape)
        for mesh_dim, p in enumerate(self.placements):
            idx = coord[mesh_dim]
            w = self.mesh.size(mesh_dim)
            if isinstance(p, Shard) and not isinstance(p, InterleavedShard):
                total = shape[p.dim]
                offsets[p.dim] += Shard.chunk_offset(total, w, idx)
                shape[p.dim] = Shard.chunk_size(total, w, idx)
        return tuple(offsets)


def make_spec(
    mesh: DeviceMesh,
    placements: Sequence[Placement],
    tensor: Optional[torch.Tensor] = None,
    *,
    shape: Optional[Sequence[int]] = None,
    stride: Optional[Sequence[int]] = None,
    dtype: Optional[torch.dtype] = None,
) -> DTensorSpec:
    tm = None
    if tensor is not None:
        tm = TensorMeta(tensor.shape, tuple(tensor.stride()), tensor.dtype)
    elif shape is not None:
        if stride is None:
            st, acc = [], 1
            for s in reversed(list(shape)):
                st.append(acc)
                acc *= s
            stride = tuple(reversed(st))
        tm = TensorMeta(torch.Size(shape), tuple(stride), dtype or torch.float32)
    return DTensorSpec(mesh, tuple(placements), tm)
