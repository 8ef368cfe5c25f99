from .api import parallelize_experts
from .experts_allocator import BasicExpertsAllocator, ExpertsAllocator
from .token_dispatcher import BasicTokenDispatcher, TokenDispatcher
from ._utils import global_all_to_all_single

__all__ = [
    "parallelize_experts",
    "ExpertsAllocator",
    "BasicExpertsAllocator",
    "TokenDispatcher",
    "BasicTokenDispatcher",
    "global_all_to_all_single",
]
