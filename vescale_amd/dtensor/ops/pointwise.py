"""Pointwise op registrations (parity: legacy/vescale/dtensor/ops/
pointwise_ops.py, vescale/dtensor/_ops/_pointwise_ops.py)."""
from __future__ import annotations

import torch

from .common import pointwise_rule

aten = torch.ops.aten

POINTWISE_OPS = [
    aten.abs, aten.abs_, aten.add, aten.add_, aten.addcdiv, aten.addcdiv_,
    aten.addcmul, aten.addcmul_, aten.atan, aten.atan2, aten.bitwise_and,
    aten.bitwise_and_, aten.bitwise_not, aten.bitwise_or, aten.bitwise_or_,
    aten.bitwise_xor, aten.ceil, aten.ceil_, aten.clamp, aten.clamp_,
    aten.clamp_min, aten.clamp_min_, aten.clamp_max, aten.clamp_max_,
    aten.cos, aten.cos_, aten.cosh, aten.div, aten.div_, aten.elu, aten.elu_,
    aten.eq, aten.eq_, aten.erf, aten.erf_, aten.exp, aten.exp_, aten.expm1,
    aten.floor, aten.floor_, aten.floor_divide, aten.fmod, aten.frac,
    aten.ge, aten.ge_, aten.gelu, aten.gt, aten.gt_, aten.hardtanh,
    aten.hardtanh_, aten.isinf, aten.isnan, aten.le, aten.le_, aten.leaky_relu,
    aten.leaky_relu_, aten.lerp, aten.lerp_, aten.log, aten.log_, aten.log10,
    aten.log1p, aten.log2, aten.logical_and, aten.logical_not, aten.logical_or,
    aten.lt, aten.lt_, aten.maximum, aten.minimum, aten.mul, aten.mul_,
    aten.ne, aten.ne_, aten.neg, aten.neg_, aten.pow, aten.pow_,
    aten.reciprocal, aten.relu, aten.relu_, aten.remainder, aten.round,
    aten.rsqrt, aten.rsqrt_, aten.rsub, aten.sigmoid, aten.sigmoid_,
    aten.sign, aten.silu, aten.silu_, aten.sin, aten.sin_, aten.sinh,
    aten.sqrt, aten.sqrt_, aten.sub, aten.sub_, aten.tan, aten.tanh,
    aten.tanh_, aten.threshold, aten.trunc, aten.square, aten.square_,
    aten.logit, aten.polar, aten.hypot, aten.nan_to_num, aten.nan_to_num_,
    aten.masked_fill, aten.masked_fill_, aten.where, aten.fill, aten.fill_,
    aten.zero_, aten.tril, aten.triu, aten.tril_, aten.triu_,
    aten.clamp_min, aten.sgn, aten.exponential_,
    # backward pointwise
    aten.gelu_backward, aten.silu_backward, aten.sigmoid_backward,
    aten.tanh_backward, aten.threshold_backward, aten.elu_backward,
    aten.leaky_relu_backward, aten.hardtanh_backward, aten.logit_backward,
    # casts / copies with same layout
    aten._to_copy, aten.copy_, aten.to,
]


def register(dispatcher):
    for op in POINTWISE_OPS:
        try:
            dispatcher.register_rule(op, pointwise_rule)
        except AttributeError:
            pass

    # foreach ops: dispatcher handler unwraps element-wise (see tensor_ops
    # register for the handler); the plain rule also works per-element when
    # placements align, so register the common ones through a handler there.
