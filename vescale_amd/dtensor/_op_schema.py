"""Op schema objects passed to sharding rules.

Parity concept: legacy/vescale/dtensor/op_schema.py + sharding_prop tables.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import torch

from ._dtensor_spec import DTensorSpec
from .placement_types import Placement


@dataclass
class OpSchema:
    """Flattened view of an aten call with DTensorSpec standing in for every
    DTensor argument."""

    op: torch._ops.OpOverload
    args_schema: Tuple[Any, ...]
    kwargs_schema: Dict[str, Any]

    @property
    def specs(self) -> List[DTensorSpec]:
        out = []

        def walk(x):
            if isinstance(x, DTensorSpec):
                out.append(x)
            elif isinstance(x, (list, tuple)):
                for y in x:
                    walk(y)

        for a in self.args_schema:
            walk(a)
        for a in self.kwargs_schema.values():
            walk(a)
        return out

    @property
    def mesh(self):
        for s in self.specs:
            return s.mesh
        raise RuntimeError("no DTensor args")


@dataclass
class OutputSharding:
    """Result of sharding propagation.

    output_spec: a DTensorSpec per tensor output (None entries for
      non-tensor / passthrough outputs); a bare DTensorSpec for single-output
      ops.
    input_targets: desired placements for each DTensor input, in the order
      OpSchema.specs enumerates them.  None = leave as-is.
    """

    output_spec: Union[DTensorSpec, Sequence[Optional[DTensorSpec]], None]
    input_targets: Optional[List[Optional[Tuple[Placement, ...]]]] = None
