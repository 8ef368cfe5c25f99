"""DeviceMesh — the communicator bootstrap for the MI355X framework.

One process per GPU; `torch.distributed` backend "nccl" is RCCL on ROCm and
rides xGMI intra-node.  Capability parity with reference
`legacy/vescale/dtensor/device_mesh.py` (construct from ranks array or an
existing ProcessGroup, per-dim PG creation, submesh slicing by name, auto
set_device, mesh validation) — re-designed, not translated.
"""
from __future__ import annotations

import logging
import os
import threading
from typing import Dict, List, Optional, Sequence, Tuple, Union

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

__all__ = ["DeviceMesh", "init_device_mesh"]


class _MeshEnv(threading.local):
    def __init__(self) -> None:
        self.mesh_stack: List["DeviceMesh"] = []
        # child mesh -> (parent mesh, dim in parent)
        self.child_to_parent: Dict["DeviceMesh", Tuple["DeviceMesh", int]] = {}

    def create_child_mesh(self, parent: "DeviceMesh", mesh_dim: int, name: str) -> "DeviceMesh":
        # slice out the 1-D submesh containing this rank along mesh_dim
        cur_rank = parent.get_rank()
        pg = parent.get_group(mesh_dim)
        # compute the ranks in this rank's submesh along mesh_dim
        mesh_nd = parent.mesh
        # move target dim last, flatten others
        perm = [d for d in range(mesh_nd.ndim) if d != mesh_dim] + [mesh_dim]
        rows = mesh_nd.permute(perm).reshape(-1, mesh_nd.size(mesh_dim))
        my_row = None
        for r in rows:
            if cur_rank in r.tolist():
                my_row = r
                break
        assert my_row is not None, f"rank {cur_rank} not in mesh {mesh_nd}"
        child = DeviceMesh(
            parent.device_type,
            my_row,
            mesh_dim_names=(name,),
            _init_process_groups=False,
        )
        child._dim_groups = [pg]
        _LIVE_MESHES.setdefault(child._registry_key(), child)
        self.child_to_parent[child] = (parent, mesh_dim)
        return child


_mesh_env = _MeshEnv()

# live meshes by description, for pickle rebinding: a mesh travels as its
# description (device_type + rank array + names) and re-attaches to the
# process's live communicators on load — ProcessGroups never serialize.
# (Reference parity: dmodule/test_saveload.py torch.saves DTensor state
# dicts; DCP checkpointing does the same via spec metadata.)
_LIVE_MESHES: Dict[tuple, "DeviceMesh"] = {}


def _get_device_handle(device_type: str):
    return getattr(torch, device_type, None) if device_type != "cpu" else None


class DeviceMesh:
    """An n-dim array of global ranks + one ProcessGroup per mesh dim.

    mesh dims are ordered outermost-first (dim 0 varies slowest).
    """

    device_type: str
    mesh: torch.Tensor

    def __init__(
        self,
        device_type: str,
        mesh: Union[torch.Tensor, Sequence[int], Sequence[Sequence[int]]],
        *,
        mesh_dim_names: Optional[Tuple[str, ...]] = None,
        pg: Optional[dist.ProcessGroup] = None,
        _init_process_groups: bool = True,
    ) -> None:
        self.device_type = device_type
        self.mesh = (
            mesh.detach().cpu()
            if isinstance(mesh, torch.Tensor)
            else torch.tensor(mesh, dtype=torch.int64)
        )
        if self.mesh.ndim == 0:
            self.mesh = self.mesh.reshape(1)
        self.mesh_dim_names = tuple(mesh_dim_names) if mesh_dim_names else None
        if self.mesh_dim_names:
            assert len(self.mesh_dim_names) == self.mesh.ndim
        self._dim_groups: List[dist.ProcessGroup] = []
        self._flat_rank_map = {int(r): i for i, r in enumerate(self.mesh.flatten())}

        if pg is not None:
            # build from an existing ProcessGroup (reference device_mesh.py:186-192)
            ranks = dist.get_process_group_ranks(pg)
            assert self.mesh.flatten().tolist() == ranks or mesh is None, (
                "mesh ranks must match the given process group"
            )
            assert self.mesh.ndim == 1, "pg-based construction is 1-D"
            self._dim_groups = [pg]
            _LIVE_MESHES.setdefault(self._registry_key(), self)
            self._setup_device()
            return

        if _init_process_groups:
            self._maybe_init_default_pg()
            self._setup_device()
            self._init_dim_groups()
            _LIVE_MESHES.setdefault(self._registry_key(), self)

    # ------------------------------------------------------------------
    def _maybe_init_default_pg(self) -> None:
        if not dist.is_initialized():
            backend = "nccl" if self.device_type == "cuda" else "gloo"
            dist.init_process_group(backend=backend)
        world_size = dist.get_world_size()
        if self.mesh.numel() > world_size:
            raise RuntimeError(
                f"mesh has {self.mesh.numel()} ranks but world size is {world_size}"
            )

    def _setup_device(self) -> None:
        if self.device_type == "cuda" and torch.cuda.is_available():
            # one process per GPU: pin device by LOCAL_RANK (modulo device
            # count so oversubscribed debug runs on 1 GPU still work)
            local_rank = int(os.environ.get("LOCAL_RANK", dist.get_rank()))
            torch.cuda.set_device(local_rank % torch.cuda.device_count())

    def _init_dim_groups(self) -> None:
        if self.mesh.ndim == 1 and self.mesh.numel() == dist.get_world_size():
            default = dist.group.WORLD
            self._dim_groups = [default]
            return
        for dim in range(self.mesh.ndim):
            self._dim_groups.append(self._new_group_for_dim(dim))

    def _new_group_for_dim(self, dim: int) -> dist.ProcessGroup:
        perm = [d for d in range(self.mesh.ndim) if d != dim] + [dim]
        rows = self.mesh.permute(perm).reshape(-1, self.mesh.size(dim))
        my_group = None
        cur = dist.get_rank()
        for row in rows:
            ranks = row.tolist()
            g = dist.new_group(ranks=ranks)
            if cur in ranks:
                my_group = g
        assert my_group is not None or cur not in self._flat_rank_map
        return my_group

    # ------------------------------------------------------------------
    @property
    def ndim(self) -> int:
        return self.mesh.ndim

    @property
    def shape(self) -> Tuple[int, ...]:
        return tuple(self.mesh.shape)

    def size(self, dim: Optional[int] = None) -> int:
        return self.mesh.numel() if dim is None else self.mesh.size(dim)

    def get_rank(self) -> int:
        return dist.get_rank() if dist.is_initialized() else 0

    def get_local_rank(self, mesh_dim: int = 0) -> int:
        coord = self.get_coordinate()
        assert coord is not None, "rank not in mesh"
        return coord[mesh_dim]

    def get_coordinate(self) -> Optional[List[int]]:
        """This rank's coordinate in the mesh, or None if not a participant."""
        r = self.get_rank()
        if r not in self._flat_rank_map:
            return None
        flat_idx = self._flat_rank_map[r]
        coord = []
        for s in reversed(self.mesh.shape):
            coord.append(flat_idx % s)
            flat_idx //= s
        return list(reversed(coord))

    def get_group(self, mesh_dim: Union[int, str] = 0) -> dist.ProcessGroup:
        if isinstance(mesh_dim, str):
            assert self.mesh_dim_names, "mesh has no dim names"
            mesh_dim = self.mesh_dim_names.index(mesh_dim)
        self._ensure_groups()
        return self._dim_groups[mesh_dim]

    def get_all_groups(self) -> List[dist.ProcessGroup]:
        self._ensure_groups()
        return list(self._dim_groups)

    # ------------------------------------------------------------------
    # pickling: a mesh serializes as its DESCRIPTION only (communicators
    # never pickle).  On load it rebinds to the live mesh with the same
    # description, or lazily re-creates the dim groups on first use (all
    # ranks load in lockstep in the save/load pattern, so group creation
    # stays collective-ordered).
    def _registry_key(self) -> tuple:
        return (
            self.device_type,
            tuple(self.mesh.shape),
            tuple(self.mesh.flatten().tolist()),
            self.mesh_dim_names,
        )

    def __getstate__(self):
        return {
            "device_type": self.device_type,
            "mesh": self.mesh.tolist(),
            "mesh_dim_names": self.mesh_dim_names,
        }

    def __setstate__(self, state):
        self.device_type = state["device_type"]
        self.mesh = torch.tensor(state["mesh"], dtype=torch.int64)
        if self.mesh.ndim == 0:
            self.mesh = self.mesh.reshape(1)
        self.mesh_dim_names = state["mesh_dim_names"]
        self._flat_rank_map = {int(r): i for i, r in enumerate(self.mesh.flatten())}
        live = _LIVE_MESHES.get(self._registry_key())
        self._dim_groups = live._dim_groups if live is not None else None

    def _ensure_groups(self) -> None:
        if self._dim_groups is not None:
            return
        live = _LIVE_MESHES.get(self._registry_key())
        if live is not None and live._dim_groups:
            self._dim_groups = live._dim_groups
            return
        if not dist.is_initialized():
            raise RuntimeError(
                f"unpickled {self!r} has no live communicators and "
                "torch.distributed is not initialized — construct the mesh "
                "(or init_device_mesh) before using the loaded DTensors"
            )
        self._dim_groups = []
        self._init_dim_groups()
        _LIVE_MESHES.setdefault(self._registry_key(), self)

    def get_dim_groups(self, mesh_dim: Optional[int] = None):
        if mesh_dim is None:
            return self.get_all_groups()
        return self.get_group(mesh_dim)

    # submesh slicing: mesh["TP"] -> 1-D child mesh
    def __getitem__(self, name: str) -> "DeviceMesh":
        assert self.mesh_dim_names and name in self.mesh_dim_names, (
            f"unknown mesh dim {name!r}; have {self.mesh_dim_names}"
        )
        dim = self.mesh_dim_names.index(name)
        return _mesh_env.create_child_mesh(self, dim, name)

    def __eq__(self, other) -> bool:
        if self is other:
            return True
        return (
            isinstance(other, DeviceMesh)
            and self.device_type == other.device_type
            and self.mesh.shape == other.mesh.shape
            and bool(torch.equal(self.mesh, other.mesh))
        )

    def __hash__(self) -> int:
        return hash((self.device_type, self.mesh.shape, tuple(self.mesh.flatten().tolist())))

    def __repr__(self) -> str:
        return f"DeviceMesh({self.device_type}, {self.mesh.tolist()}, names={self.mesh_dim_names})"

    # a mesh is a process-global handle to live communicators; copying one
    # is meaningless and deepcopy would die pickling ProcessGroups (dynamo
    # guard construction deepcopies DTensor specs -> mesh)
    def __copy__(self):
        return self

    def __deepcopy__(self, memo):
        return self


def init_device_mesh(
    device_type: str,
    mesh_shape: Sequence[int],
    *,
    mesh_dim_names: Optional[Sequence[str]] = None,
) -> DeviceMesh:
    """Create a DeviceMesh covering ranks [0, prod(mesh_shape))."""
    n = 1
    for s in mesh_shape:
        n *= s
    mesh = torch.arange(n, dtype=torch.int64).reshape(tuple(mesh_shape))
    return DeviceMesh(
        device_type,
        mesh,
        mesh_dim_names=tuple(mesh_dim_names) if mesh_dim_names else None,
    )
