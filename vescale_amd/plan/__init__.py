from .pipeline_parallel import PipelineParallelPlan
from .spec import (
    ModeType,
    PipelineP2PSpec,
    PipelineScheduleType,
    PipelineSplitMethodType,
    TracerType,
)

__all__ = [
    "PipelineParallelPlan",
    "ModeType",
    "PipelineScheduleType",
    "PipelineSplitMethodType",
    "PipelineP2PSpec",
    "TracerType",
]
