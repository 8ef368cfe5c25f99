"""Auto-plan (DMP): walk the module tree, match policies by class name,
assemble a DModule sharding plan, parallelize.

Parity: legacy/vescale/dmp/dmp.py:37-220 (auto_parallelize_module,
PlanGenerator, plan overriding policy).
"""
from __future__ import annotations

import re
from typing import Callable, Dict, Optional

import torch.nn as nn

from ..dmodule import parallelize_module
from ..dtensor import DeviceMesh
from .policies import REGISTRY

_override_policy: Optional[Callable] = None


def set_plan_overriding_policy(fn: Callable):
    global _override_policy
    _override_policy = fn


def get_plan_overriding_policy() -> Optional[Callable]:
    return _override_policy


class PlanGenerator:
    def __init__(self, registry: Dict[str, Callable] = None, sp: bool = True):
        self.registry = registry or REGISTRY
        self.sp = sp

    def generate(self, module: nn.Module) -> Dict[str, Dict]:
        param: Dict = {}
        fwd: Dict = {}
        matched = set()
        for fqn, mod in module.named_modules():
            cls = type(mod).__name__
            # skip submodules of an already-matched module
            if any(fqn.startswith(m + ".") for m in matched if m):
                continue
            provider = self.registry.get(cls)
            if provider is None:
                continue
            p, f = provider(re.escape(fqn) if fqn else fqn, mod, self.sp)
            param.update(p)
            fwd.update(f)
            matched.add(fqn)
        plan = {"parameter": param, "forward": fwd}
        if _override_policy is not None:
            plan = _override_policy(module, plan)
        return plan


def auto_parallelize_module(
    module: nn.Module,
    mesh: DeviceMesh,
    *,
    sp: bool = True,
    policies: Optional[Dict[str, Callable]] = None,
) -> nn.Module:
    plan = PlanGenerator(policies, sp=sp).generate(module)
    return parallelize_module(module, mesh, plan)
