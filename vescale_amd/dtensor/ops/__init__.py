"""Sharding-rule registry: register every op family on the dispatcher."""
from __future__ import annotations


def register_all(dispatcher):
    from . import (
        attention,
        conv_ops,
        embedding,
        extra_ops,
        math_ops,
        matrix,
        pointwise,
        random_ops,
        tensor_ops,
        view_ops,
    )

    pointwise.register(dispatcher)
    matrix.register(dispatcher)
    view_ops.register(dispatcher)
    math_ops.register(dispatcher)
    tensor_ops.register(dispatcher)
    embedding.register(dispatcher)
    conv_ops.register(dispatcher)
    random_ops.register(dispatcher)
    attention.register(dispatcher)
    extra_ops.register(dispatcher)
