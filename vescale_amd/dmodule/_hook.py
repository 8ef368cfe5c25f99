"""Forward hooks converting module inputs/outputs per the plan.

Parity: legacy/vescale/dmodule/_hook.py:76-272 (PreHookInput /
PostHookOutput).  Each hook converts positional tensor args:
  plain tensor -> DTensor.from_local(t, mesh, placements)
  DTensor      -> redistribute(placements)   [=> RCCL comm]
A None placement entry leaves the arg untouched.

Plans may also be DICTS keyed by name (reference _hook.py:82-90,158,
219-251): for inputs the keys are forward() parameter names (positional
or keyword), for outputs the keys are fields of a Mapping / dataclass /
HF ModelOutput return — unnamed fields pass through untouched.
"""
from __future__ import annotations

import dataclasses
import inspect
import re
from collections.abc import Mapping
from typing import Any, Dict

import torch
import torch.nn as nn

from ..dtensor import DeviceMesh, DTensor


def _convert(x, placements, mesh: DeviceMesh):
    if placements is None or not isinstance(x, torch.Tensor):
        return x
    if isinstance(x, DTensor):
        if tuple(x.placements) == tuple(placements):
            return x
        return x.redistribute(placements=list(placements))
    return DTensor.from_local(x, mesh, list(placements))


def _convert_seq(args, plan_list, mesh):
    out = []
    ti = 0
    for a in args:
        if isinstance(a, torch.Tensor):
            pl = plan_list[ti] if plan_list is not None and ti < len(plan_list) else None
            out.append(_convert(a, pl, mesh))
            ti += 1
        else:
            out.append(a)
    return tuple(out)


def install_forward_hooks(root: nn.Module, mesh: DeviceMesh, fwd_plan: Dict[str, Any]):
    input_plans: Dict[str, Any] = {}
    output_plans: Dict[str, Any] = {}
    weight_plans: Dict[str, Any] = {}
    def strip_suffix(k: str, suffix: str):
        """Remove '.suffix' or the regex-escaped '\\.suffix'."""
        base = k[: -len("." + suffix) ]
        if base.endswith("\\"):
            base = base[:-1]
        return base

    for k, v in fwd_plan.items():
        if k.endswith(".input"):
            input_plans[strip_suffix(k, "input")] = v
        elif k.endswith(".output"):
            output_plans[strip_suffix(k, "output")] = v
        elif k.endswith(".weight_placement"):
            weight_plans[strip_suffix(k, "weight_placement")] = v
        elif k in ("input", "output"):
            # root-module IO: the fqn of the root is "", so the bare
            # suffix is the natural spelling
            (input_plans if k == "input" else output_plans)[""] = v
        else:
            raise ValueError(
                f"forward-plan key {k!r} must end in .input / .output / "
                f".weight_placement (root IO: 'input' / 'output') — a bare "
                f"module fqn would be silently ignored"
            )

    def match(table, fqn):
        for pattern, v in table.items():
            if re.fullmatch(pattern, fqn):
                return v
        return None

    for mod_name, mod in root.named_modules():
        fqn = mod_name if mod_name else ""
        ip = match(input_plans, fqn) if fqn or "" in input_plans else match(input_plans, fqn)
        op = match(output_plans, fqn)
        if ip is not None:
            if isinstance(ip, Mapping):
                mod.register_forward_pre_hook(
                    _make_named_pre_hook(ip, mesh, mod), with_kwargs=True
                )
            else:
                mod.register_forward_pre_hook(_make_pre_hook(ip, mesh))
        if op is not None:
            mod.register_forward_hook(_make_post_hook(op, mesh))


def _dynamo_opaque(fn):
    """DTensor boundaries are opaque to torch.compile (eager-first compat,
    reference patch #9's scope): the hook's redistribute/from_local runs
    eagerly via a graph break instead of dynamo tracing DeviceMesh/PG
    state (whose guards are unpicklable)."""
    disable = getattr(getattr(torch, "compiler", None), "disable", None)
    return disable(fn) if disable is not None else fn


def _make_pre_hook(plan_list, mesh):
    def hook(mod, args):
        return _convert_seq(args, plan_list, mesh)

    return _dynamo_opaque(hook)


def _make_named_pre_hook(plan_dict, mesh, mod):
    """Name-keyed input plan: convert forward() arguments by parameter
    name, whether passed positionally or as keywords."""
    try:
        sig = inspect.signature(mod.forward)
    except (TypeError, ValueError):
        sig = None

    def hook(mod, args, kwargs):
        if sig is None:
            return args, kwargs
        try:
            bound = sig.bind(*args, **kwargs)
        except TypeError:
            return args, kwargs
        for name, pl in plan_dict.items():
            if name in bound.arguments:
                bound.arguments[name] = _convert(bound.arguments[name], pl, mesh)
        return bound.args, bound.kwargs

    return _dynamo_opaque(hook)


def _convert_named(output, plan_dict, mesh):
    """Convert named fields of a Mapping / dataclass / HF-ModelOutput
    return; fields absent from the plan pass through untouched."""
    if isinstance(output, Mapping):
        new = {
            k: _convert(v, plan_dict.get(k), mesh) if k in plan_dict else v
            for k, v in output.items()
        }
        return type(output)(**new)
    if dataclasses.is_dataclass(output) and not isinstance(output, type):
        for f in dataclasses.fields(output):
            if f.name in plan_dict:
                v = _convert(getattr(output, f.name), plan_dict[f.name], mesh)
                # HF ModelOutput subclasses are also dicts; plain frozen
                # dataclasses need object.__setattr__
                try:
                    setattr(output, f.name, v)
                except (AttributeError, dataclasses.FrozenInstanceError):
                    object.__setattr__(output, f.name, v)
                if isinstance(output, Mapping) and f.name in output:
                    output[f.name] = v
        return output
    return output


def _make_post_hook(plan_list, mesh):
    def hook(mod, args, output):
        if isinstance(plan_list, Mapping):
            return _convert_named(output, plan_list, mesh)
        if isinstance(output, torch.Tensor):
            pl = plan_list[0] if plan_list else None
            return _convert(output, pl, mesh)
        if isinstance(output, (tuple, list)):
            return type(output)(_convert_seq(output, plan_list, mesh))
        return output

    return _dynamo_opaque(hook)
