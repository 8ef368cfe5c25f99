"""VeDeviceMesh — the global nD strategy view used by PP/engine/checkpoint.

Parity: legacy/vescale/devicemesh_api/api.py:28-427 (init_device_mesh,
strategy ranks/coords, per-dim sub-meshes, stage predicates, dim groups).
A process-global singleton (VESCALE_DEVICE_MESH) mirrors the reference's
usage pattern.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

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
