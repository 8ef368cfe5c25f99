"""Axis-aligned box decomposition of flat (ragged) shard ranges.

Parity: vescale/dtensor/vescale_utils/checkpoint.py:70-172
(_break_ragged_box — start/middle/end box walk): a RaggedShard stores a
contiguous row-major flat range [a, b) of an N-D tensor; DCP chunks are
N-D boxes, so the range is decomposed into <= 2*ndim-1 boxes, each of
which is CONTIGUOUS in flat order (so the local bytes for box i are a
simple sub-slice of the rank's flat buffer).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

Box = Tuple[Tuple[int, ...], Tuple[int, ...]]  # (offsets, sizes)


def break_ragged_box(shape: Sequence[int], a: int, b: int) -> List[Box]:
    """Boxes covering flat range [a, b) of a row-major tensor `shape`,
    in flat order."""
    shape = tuple(shape)
    if a >= b:
        return []
    if len(shape) == 0:
        return [((), ())] if b > a else []
    if len(shape) == 1:
        return [((a,), (b - a,))]
    inner = 1
    for s in shape[1:]:
        inner *= s
    r0, c0 = divmod(a, inner)
    r1, c1 = divmod(b, inner)
    out: List[Box] = []
    if c0 != 0:
        # head partial row
        end = inner if r0 < r1 else c1
        for o, s in break_ragged_box(shape[1:], c0, end):
            out.append(((r0,) + o, (1,) + s))
        r0 += 1
        if r0 > r1:
            return out
    if r1 > r0:
        out.append(((r0,) + (0,) * (len(shape) - 1), (r1 - r0,) + shape[1:]))
    if c1 != 0 and r1 >= r0:
        for o, s in break_ragged_box(shape[1:], 0, c1):
            out.append(((r1,) + o, (1,) + s))
    return out


def box_numel(box: Box) -> int:
    n = 1
    for s in box[1]:
        n *= s
    return n


def box_flat_start(shape: Sequence[int], box: Box) -> int:
    """Flat index of the box's first element (boxes from break_ragged_box
    are flat-contiguous)."""
    off = 0
    stride = 1
    for d in range(len(shape) - 1, -1, -1):
        off += box[0][d] * stride
        stride *= shape[d]
    return off
