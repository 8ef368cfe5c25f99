"""Random-op rules: run under the RNG tracker region so sharded results can
reproduce single-device randomness bitwise (reference dispatch.py:310-320 +
random.py; the bitwise path on GPU uses our sharded-philox HIP kernels)."""
from __future__ import annotations

import torch

from .._op_schema import OpSchema, OutputSharding
from ..placement_types import Partial, Replicate
from .common import out_spec

aten = torch.ops.aten

RANDOM_OPS = [
    aten.native_dropout.default,
    aten.uniform_.default,
    aten.normal_.default,
    aten.rand_like.default,
    aten.randn_like.default,
    aten.randint_like,
    aten.bernoulli.default,
    aten.bernoulli_.float,
]


def dropout_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    pl = [Replicate() if isinstance(p, Partial) else p for p in s.placements]
    targets = [tuple(pl)]
    o0 = out_spec(s.mesh, pl, tuple(s.shape), s.dtype)
    o1 = out_spec(s.mesh, pl, tuple(s.shape), torch.bool)
    return OutputSharding([o0, o1], targets)


def inplace_random_rule(schema: OpSchema) -> OutputSharding:
    s = schema.specs[0]
    return OutputSharding(out_spec(s.mesh, s.placements, tuple(s.shape), s.dtype), None)


def register(dispatcher):
    dispatcher.register_rule(aten.native_dropout.default, dropout_rule)
    for op in (
        aten.uniform_.default,
        aten.normal_.default,
        aten.rand_like.default,
        aten.randn_like.default,
        aten.bernoulli.default,
    ):
        dispatcher.register_rule(op, inplace_random_rule)
    for op in RANDOM_OPS:
        try:
            dispatcher.register_random(op)
        except Exception:
            pass
    # dropout backward is pointwise
    from .common import pointwise_rule

    dispatcher.register_rule(aten.native_dropout_backward.default, pointwise_rule)
