from .api import is_experts_parallized, parallelize_experts, rebalance_experts
from .experts_allocator import (
    BasicExpertsAllocator,
    ExpertsAllocator,
    LoadBalancedExpertsAllocator,
)
from .token_dispatcher import BasicTokenDispatcher, TokenDispatcher
from ._utils import global_all_to_all_single

__all__ = [
    "parallelize_experts",
    "rebalance_experts",
    "LoadBalancedExpertsAllocator",
    "ExpertsAllocator",
    "BasicExpertsAllocator",
    "TokenDispatcher",
    "BasicTokenDispatcher",
    "global_all_to_all_single",
]
