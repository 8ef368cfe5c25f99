"""Factory dispatch mode: torch.zeros/ones/empty/full/arange called INSIDE
a DModule forward produce DTensors on the module's mesh.

Parity: legacy/vescale/dmodule/_factory.py:57 (FactoryDispatchMode wired by
DModule.prepare_factory, _dmodule.py:389).
"""
from __future__ import annotations


import torch
from torch.overrides import TorchFunctionMode

from ..dtensor import DeviceMesh, DTensor, Replicate, distribute_tensor

_FACTORIES = {
    torch.zeros, torch.ones, torch.empty, torch.full, torch.rand, torch.randn,
    torch.arange,
}


class FactoryDispatchMode(TorchFunctionMode):
    def __init__(self, mesh: DeviceMesh, placements=None):
        super().__init__()
        self.mesh = mesh
        self.placements = placements

    def __torch_function__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        if func in _FACTORIES and "out" not in kwargs:
            t = func(*args, **kwargs)
            pl = self.placements or [Replicate()] * self.mesh.ndim
            return distribute_tensor(t, self.mesh, pl)
        return func(*args, **kwargs)


def install_factory_mode(module, mesh: DeviceMesh, placements=None):
    """Wrap the module's forward so factory calls inside it emit DTensors
    (reference prepare_factory)."""
    orig_forward = module.forward

    def forward(*args, **kwargs):
        with FactoryDispatchMode(mesh, placements):
            return orig_forward(*args, **kwargs)

    module.forward = forward
    return module
