"""PipelineParallelPlan dataclass (parity:
legacy/vescale/plan/pipeline_parallel.py:28)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, List, Optional, Sequence

from .spec import (
    ModeType,
    PipelineScheduleType,
    PipelineSplitMethodType,
    TracerType,
)


@dataclass
class PipelineParallelPlan:
    mode: ModeType = ModeType.MANUAL_EAGER
    split_method: PipelineSplitMethodType = PipelineSplitMethodType.MANUAL
    schedule_type: PipelineScheduleType = PipelineScheduleType.SIMPLE_1F1B
    num_stages: int = 2
    virtual_chunks: int = 1
    smallest_split_units: List[str] = field(default_factory=list)
    split_points: List[str] = field(default_factory=list)
    batch_p2p_comm: bool = True
    overlap_p2p_comm: bool = False
    use_zero_bubble: bool = False
    tracer_type: TracerType = TracerType.MANUAL
    shared_modules: List[List[str]] = field(default_factory=list)
    # shapes for p2p placeholder allocation (dynamic handshake used if None)
    p2p_tensor_shape: Optional[Sequence[int]] = None
    p2p_tensor_dtype: Any = None
    reuse_p2p_tensor_shape: bool = False
