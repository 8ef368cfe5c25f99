"""VeDeviceMesh — the global nD strategy view used by PP/engine/checkpoint.

Parity: legacy/vescale/devicemesh_api/api.py:28-427 (init_device_mesh,
strategy ranks/coords, per-dim sub-meshes, stage predicates, dim groups).
A process-global singleton (VESCALE_DEVICE_MESH) mirrors the reference's
usage pattern.
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ..dtensor.device_mesh import DeviceMesh


class VeDeviceMesh:
    PP_DIM_NAMES = ("PP", "pp", "pipe", "pipeline")
    DP_DIM_NAMES = ("DP", "dp", "data")
    TP_DIM_NAMES = ("TP", "tp", "tensor", "model")

    def __init__(self):
        self._mesh: Optional[DeviceMesh] = None
        self._dim_names: Tuple[str, ...] = ()

    # ------------------------------------------------------------------
    def init_device_mesh(
        self,
        device_type: str,
        mesh_shape: Sequence[int],
        *,
        mesh_dim_names: Optional[Sequence[str]] = None,
        check_uniqueness: bool = False,
    ) -> DeviceMesh:
        if check_uniqueness and self._mesh is not None:
            raise RuntimeError("global VeDeviceMesh already initialized")
        names = tuple(mesh_dim_names) if mesh_dim_names else tuple(
            f"dim{i}" for i in range(len(mesh_shape))
        )
        n = 1
        for s in mesh_shape:
            n *= s
        mesh = DeviceMesh(
            device_type,
            torch.arange(n, dtype=torch.int64).reshape(tuple(mesh_shape)),
            mesh_dim_names=names,
        )
        self._mesh = mesh
        self._dim_names = names
        return mesh

    def get(self) -> Optional[DeviceMesh]:
        return self._mesh

    @property
    def ndim(self) -> int:
        return self._mesh.ndim

    # ------------------------------------------------------------------
    def _dim_index(self, aliases: Tuple[str, ...]) -> Optional[int]:
        for i, name in enumerate(self._dim_names):
            if name in aliases:
                return i
        return None

    def _rank_on(self, aliases, default: int = 0) -> int:
        idx = self._dim_index(aliases)
        if idx is None:
            return default
        coord = self._mesh.get_coordinate()
        return coord[idx] if coord is not None else default

    def _size_on(self, aliases, default: int = 1) -> int:
        idx = self._dim_index(aliases)
        return self._mesh.size(idx) if idx is not None else default

    def get_pipeline_parallel_rank(self) -> int:
        return self._rank_on(self.PP_DIM_NAMES)

    def get_data_parallel_rank(self) -> int:
        return self._rank_on(self.DP_DIM_NAMES)

    def get_tensor_parallel_rank(self) -> int:
        return self._rank_on(self.TP_DIM_NAMES)

    def get_pipeline_parallel_world_size(self) -> int:
        return self._size_on(self.PP_DIM_NAMES)

    def get_data_parallel_world_size(self) -> int:
        return self._size_on(self.DP_DIM_NAMES)

    def get_tensor_parallel_world_size(self) -> int:
        return self._size_on(self.TP_DIM_NAMES)

    # sub-meshes ---------------------------------------------------------
    def get_tensor_parallel_mesh(self) -> DeviceMesh:
        return self._mesh[self._dim_names[self._dim_index(self.TP_DIM_NAMES)]]

    def get_data_parallel_mesh(self) -> DeviceMesh:
        return self._mesh[self._dim_names[self._dim_index(self.DP_DIM_NAMES)]]

    def get_pipeline_parallel_mesh(self) -> DeviceMesh:
        return self._mesh[self._dim_names[self._dim_index(self.PP_DIM_NAMES)]]

    def get_data_parallel_dim_groups(self):
        idx = self._dim_index(self.DP_DIM_NAMES)
        return self._mesh.get_group(idx) if idx is not None else None

    def get_tensor_parallel_dim_groups(self):
        idx = self._dim_index(self.TP_DIM_NAMES)
        return self._mesh.get_group(idx) if idx is not None else None

    # stage predicates ---------------------------------------------------
    def is_first_stage(self) -> bool:
        return self.get_pipeline_parallel_rank() == 0

    def is_last_stage(self) -> bool:
        return self.get_pipeline_parallel_rank() == self.get_pipeline_parallel_world_size() - 1

    # strategy-coordinate queries (reference api.py:188-281) --------------
    def get_strategy_coordinate(self, local_rank: Optional[int] = None):
        """Coordinate of `local_rank` (default: this rank) over the mesh's
        strategy dimensions — unlike DeviceMesh.get_coordinate(), any
        rank's coordinate can be queried (reference api.py:188)."""
        m = self._require().mesh
        r = dist.get_rank() if local_rank is None else local_rank
        return [int(i) for i in (m == r).nonzero(as_tuple=True)]

    def lookup_rank(self, dim) -> int:
        """This rank's strategy index along `dim` (name or position)."""
        coord = self.get_strategy_coordinate()
        if isinstance(dim, str):
            return coord[self._dim_names.index(dim)]
        return coord[dim]

    def get_strategy_size(self, dim) -> int:
        """Size of strategy dimension `dim` (name or position)."""
        if isinstance(dim, str):
            return self._require().size(self._dim_names.index(dim))
        return self._require().size(dim)

    def get_local_rank(self) -> int:
        """Rank within this machine (node-local device index)."""
        import torch

        per_node = (
            torch.cuda.device_count() if torch.cuda.is_available() else 8
        )
        return dist.get_rank() % max(per_node, 1)

    def get_coordinate(self):
        return self._require().get_coordinate()

    def size(self, dim: Optional[int] = None) -> int:
        return self._require().size(dim)

    @property
    def shape(self):
        return self._require().shape

    def __getitem__(self, name: str) -> DeviceMesh:
        return self._require()[name]

    def get_global_tensor_parallel_meshes(self):
        """Rank rows of every TP submesh (reference api.py:361): the global
        view a scheduler uses to enumerate TP groups."""
        idx = self._dim_index(self.TP_DIM_NAMES)
        assert idx is not None
        m = self._require().mesh
        perm = [d for d in range(m.ndim) if d != idx] + [idx]
        return [row.tolist() for row in m.permute(perm).reshape(-1, m.size(idx))]

    def get_global_pipeline_parallel_meshes(self):
        """Rank rows of every PP submesh (reference api.py:333)."""
        idx = self._dim_index(self.PP_DIM_NAMES)
        assert idx is not None
        m = self._require().mesh
        perm = [d for d in range(m.ndim) if d != idx] + [idx]
        return [row.tolist() for row in m.permute(perm).reshape(-1, m.size(idx))]

    def _require(self) -> DeviceMesh:
        mesh = self.get()
        assert mesh is not None, "Must initialize global DeviceMesh first!"
        return mesh

    # stage peers --------------------------------------------------------
    def get_global_rank_of_stage(self, stage: int) -> int:
        """Global rank holding `stage` at this rank's (dp, tp) coordinate."""
        idx = self._dim_index(self.PP_DIM_NAMES)
        assert idx is not None
        coord = list(self._mesh.get_coordinate())
        coord[idx] = stage
        m = self._mesh.mesh
        for c in coord:
            m = m[c]
        return int(m)


VESCALE_DEVICE_MESH = VeDeviceMesh()
