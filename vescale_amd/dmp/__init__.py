from .dmp import PlanGenerator, auto_parallelize_module, get_plan_overriding_policy, set_plan_overriding_policy
from .policies import REGISTRY, register_policy

__all__ = [
    "auto_parallelize_module",
    "PlanGenerator",
    "register_policy",
    "REGISTRY",
    "set_plan_overriding_policy",
    "get_plan_overriding_policy",
]
