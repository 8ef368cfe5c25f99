"""CommDebugMode — count collectives under a context manager.

Parity: vescale/dtensor/debug/_comm_mode.py:1-103.  Counts mesh-collective
invocations by monkey-wrapping vescale_amd.dtensor._collective_utils for
the scope of the context.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict

from ..dtensor import _collective_utils as cc

_TRACKED = [
    "mesh_all_reduce",
    "mesh_all_gather",
    "mesh_reduce_scatter",
    "mesh_broadcast",
    "mesh_scatter",
    "mesh_scatter_ragged",
    "mesh_all_to_all",
    "mesh_all_to_all_single",
]


class CommDebugMode:
    def __init__(self):
        self.comm_counts: Dict[str, int] = defaultdict(int)
        self._orig = {}

    def get_comm_counts(self) -> Dict[str, int]:
        return dict(self.comm_counts)

    @property
    def total(self) -> int:
        return sum(self.comm_counts.values())

    def __enter__(self):
        for name in _TRACKED:
            orig = getattr(cc, name)
            self._orig[name] = orig

            def make(nm, fn):
                def wrapper(*a, **k):
                    # meta tensors short-circuit inside the helper (no c10d
                    # call happens) — count only real communications
                    import torch as _t

                    if not any(
                        isinstance(x, _t.Tensor) and x.device.type == "meta"
                        for x in a
                    ):
                        self.comm_counts[nm] += 1
                    return fn(*a, **k)

                return wrapper

            setattr(cc, name, make(name, orig))
        return self

    def __exit__(self, *exc):
        for name, fn in self._orig.items():
            setattr(cc, name, fn)
        return False
