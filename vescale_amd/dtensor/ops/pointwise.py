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
    aten.zero_,
    aten.clamp_min, aten.sgn, aten.exponential_,
    # backward pointwise
    aten.gelu_backward, aten.silu_backward, aten.sigmoid_backward,
    aten.tanh_backward, aten.threshold_backward, aten.elu_backward,
    aten.leaky_relu_backward, aten.hardtanh_backward, aten.logit_backward,
    # casts / copies with same layout
    aten._to_copy, aten.copy_, aten.to,
    # breadth sweep (reference registers these too —
    # all plain element-local ops, same pointwise rule)
    aten.acos, aten.acos_, aten.acosh, aten.acosh_, aten.asin, aten.asin_,
    aten.asinh, aten.asinh_, aten.atan_, aten.atan2_, aten.atanh,
    aten.atanh_, aten.bitwise_left_shift, aten.bitwise_left_shift_,
    aten.bitwise_not_, aten.bitwise_right_shift, aten.bitwise_right_shift_,
    aten.bitwise_xor_, aten.clip, aten.clip_, aten.conj_physical,
    aten.conj_physical_, aten.copysign, aten.copysign_, aten.cosh_,
    aten.deg2rad, aten.deg2rad_, aten.digamma, aten.digamma_, aten.erfc,
    aten.erfc_, aten.erfinv, aten.erfinv_, aten.exp2, aten.exp2_,
    aten.expm1_, aten.float_power, aten.float_power_, aten.floor_divide_,
    aten.fmod_, aten.frac_, aten.hypot_, aten.i0, aten.i0_, aten.igamma,
    aten.igamma_, aten.igammac, aten.igammac_, aten.isneginf, aten.isposinf,
    aten.ldexp, aten.ldexp_, aten.lgamma, aten.lgamma_, aten.log10_,
    aten.log1p_, aten.log2_, aten.logaddexp, aten.logaddexp2,
    aten.logical_and_, aten.logical_not_, aten.logical_or_,
    aten.logical_xor, aten.logical_xor_, aten.logit_, aten.mvlgamma,
    aten.mvlgamma_, aten.nextafter, aten.nextafter_, aten.positive,
    aten.rad2deg, aten.rad2deg_, aten.reciprocal_, aten.remainder_,
    aten.round_, aten.sgn_, aten.sign_, aten.signbit, aten.sinc, aten.sinc_,
    aten.sinh_, aten.tan_, aten.true_divide, aten.trunc_, aten.xlogy,
    aten.xlogy_, aten._conj,
    aten.__lshift__, aten.__rshift__, aten.__ilshift__, aten.__irshift__,
    # mse loss fwd/bwd are element-local given matching shapes (reduction
    # handled by the math-ops mean/sum rules when decomposed)
    aten.mse_loss_backward,
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
