"""DModule — module-level tensor/sequence parallelism via sharding plans.

Parity: legacy/vescale/dmodule/api.py:33 parallelize_module +
_dmodule.py (regex-fqn plans, param distribution, forward hooks, Partial
grad sync).  The plan format matches the reference:

    plan = {
        "parameter": {fqn_regex: [Placement, ...] | PlacementsInterface},
        "forward":   {fqn_regex + ".input"/".output": [[Placement,...], ...]},
    }

Every redistribution the hooks trigger lowers to RCCL over xGMI through
the DTensor runtime.  Megatron-SP = activations Shard(1) between TP
regions; the Partial grads of Replicate weights are synced by the
bucketed allreduce in _grad_sync.py.
"""
from __future__ import annotations

import re
import os
from dataclasses import dataclass
from typing import Any, Dict, Optional, Sequence, Union

import torch
import torch.nn as nn

from ..dtensor import DeviceMesh, DTensor, Placement, Replicate, distribute_tensor
from ._grad_sync import install_grad_sync_methods
from ._hook import install_forward_hooks


@dataclass
class PlacementsInterface:
    """Placement spec + conversion flags (reference PlacementsInterface:
    placements/run_check/grad — async_op/defer_reshard are not needed
    here: redistribute is eager, and Partial arithmetic stays deferred by
    the default dispatch rules)."""

    placements: Sequence[Placement]
    run_check: bool = False
    # if the local tensor is ALREADY laid out as a shard, wrap with
    # from_local instead of slicing a replicated global tensor
    is_local: bool = False
    # placement to ENFORCE on this parameter's gradient (a hook
    # redistributes the incoming grad; reference PostHookGrad)
    grad: Optional[Sequence[Placement]] = None

    @classmethod
    def normalize(cls, v) -> "PlacementsInterface":
        if isinstance(v, PlacementsInterface):
            return v
        return cls(placements=tuple(v))


def _install_grad_placement_hook(param: nn.Parameter, grad_placements):
    """Enforce a placement on the param's incoming gradient (reference
    PostHookGrad): e.g. keep an SP weight grad Partial for the bucketed
    sync, or force Replicate for an immediately-consumed grad."""

    def hook(g):
        if isinstance(g, DTensor) and tuple(g.placements) != tuple(grad_placements):
            return g.redistribute(placements=list(grad_placements))
        return g

    param.register_hook(hook)


def _match_plan(fqn: str, plan: Dict[str, Any]):
    for pattern, v in plan.items():
        if re.fullmatch(pattern, fqn):
            return v
    return None


def parallelize_module(
    module: nn.Module,
    device_mesh: DeviceMesh,
    sharding_plan: Dict[str, Dict[str, Any]],
    *,
    grad_sync: Union[bool, Dict] = True,
) -> nn.Module:
    param_plan = dict(sharding_plan.get("parameter", {}))
    fwd_plan = dict(sharding_plan.get("forward", {}))

    # ---- distribute parameters ----------------------------------------
    replaced: Dict[int, nn.Parameter] = {}  # id(old) -> new (keeps weight ties)
    for mod_name, mod in module.named_modules():
        for pname, p in list(mod.named_parameters(recurse=False)):
            if id(p) in replaced:
                mod._parameters[pname] = replaced[id(p)]
                continue
            fqn = f"{mod_name}.{pname}" if mod_name else pname
            spec = _match_plan(fqn, param_plan)
            if spec is None:
                spec = PlacementsInterface([Replicate()] * device_mesh.ndim)
            else:
                spec = PlacementsInterface.normalize(spec)
            if isinstance(p.data, DTensor):
                continue
            if spec.is_local:
                d = DTensor.from_local(
                    p.data, device_mesh, spec.placements,
                    run_check=spec.run_check
                    and not os.environ.get("VESCALE_DISABLE_RUN_CHECK")
                )
            else:
                d = distribute_tensor(p.data, device_mesh, spec.placements)
            newp = nn.Parameter(d, requires_grad=p.requires_grad)
            if spec.grad is not None:
                _install_grad_placement_hook(newp, tuple(spec.grad))
            replaced[id(p)] = newp
            mod._parameters[pname] = newp
        for bname, b in list(mod.named_buffers(recurse=False)):
            fqn = f"{mod_name}.{bname}" if mod_name else bname
            spec = _match_plan(fqn, param_plan)
            if spec is not None:
                spec = PlacementsInterface.normalize(spec)
                mod._buffers[bname] = distribute_tensor(b, device_mesh, spec.placements)

    # ---- forward hooks -------------------------------------------------
    install_forward_hooks(module, device_mesh, fwd_plan)

    # ---- grad sync -----------------------------------------------------
    install_grad_sync_methods(module, device_mesh, grad_sync)

    module._is_dmodule = True
    module._device_mesh = device_mesh
    return module


def is_dmodule(module: nn.Module) -> bool:
    """True if `module` was parallelized by parallelize_module
    (reference dmodule/api.py is_dmodule)."""
    return bool(getattr(module, "_is_dmodule", False))
