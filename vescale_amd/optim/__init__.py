from .base_optimizer import BasicOptimizer
from .clip_grads import clip_grad_norm_fp32
from .distributed_optimizer import DistributedOptimizer, OptimizerStateSpec

__all__ = [
    "BasicOptimizer",
    "DistributedOptimizer",
    "OptimizerStateSpec",
    "clip_grad_norm_fp32",
]
