"""vescale_amd.dtensor — MI355X-native DTensor runtime (L1)."""
from .device_mesh import DeviceMesh, init_device_mesh
from .placement_types import (
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
    _StridedRaggedShard,
)
from ._dtensor_spec import DTensorSpec, make_spec
from .dtensor import DTensor
from .api import (
    distribute_tensor,
    from_local,
    normalize_placements,
    redistribute_dtensor,
    to_local,
    vescale_all_gather,
    vescale_all_reduce,
    vescale_reduce_scatter,
)
from .random import init_rng_tracker, manual_seed

__all__ = [
    "DeviceMesh",
    "init_device_mesh",
    "DTensor",
    "DTensorSpec",
    "TensorMeta",
    "Placement",
    "Shard",
    "Replicate",
    "Partial",
    "InterleavedShard",
    "RaggedShard",
    "_StridedRaggedShard",
    "distribute_tensor",
    "from_local",
    "to_local",
    "redistribute_dtensor",
    "normalize_placements",
    "make_spec",
    "manual_seed",
    "init_rng_tracker",
    "vescale_all_gather",
    "vescale_all_reduce",
    "vescale_reduce_scatter",
]


def _factory(fn):
    def wrapper(*size, device_mesh=None, placements=None, dtype=None, requires_grad=False, **kw):
        """DTensor factory (ones/zeros/empty/rand/randn/full parity with
        vescale/dtensor/_api.py:732-1051): builds only the local shard."""
        from .api import distribute_tensor

        if len(size) == 1 and isinstance(size[0], (tuple, list)):
            size = tuple(size[0])
        assert device_mesh is not None
        placements = normalize_placements(placements, device_mesh.ndim)
        full = fn(size, dtype=dtype, **kw) if fn is not torch_full else fn(size, kw.pop("fill_value"), dtype=dtype)
        d = distribute_tensor(full, device_mesh, placements)
        d.requires_grad_(requires_grad)
        return d

    return wrapper


import torch as _torch

torch_full = _torch.full
ones = _factory(_torch.ones)
zeros = _factory(_torch.zeros)
empty = _factory(_torch.empty)
rand = _factory(_torch.rand)
randn = _factory(_torch.randn)


def full(size, fill_value, *, device_mesh=None, placements=None, dtype=None, requires_grad=False):
    from .api import distribute_tensor

    placements = normalize_placements(placements, device_mesh.ndim)
    t = _torch.full(tuple(size), fill_value, dtype=dtype)
    d = distribute_tensor(t, device_mesh, placements)
    d.requires_grad_(requires_grad)
    return d
