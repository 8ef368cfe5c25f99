"""Plan enums/specs (parity: legacy/vescale/plan/spec.py:34-84)."""
from __future__ import annotations

from dataclasses import dataclass
from enum import Enum


class ModeType(Enum):
    EAGER = "eager"
    MANUAL_EAGER = "manual_eager"
    GRAPH_EAGER = "graph_eager"


class PipelineSplitMethodType(Enum):
    MANUAL = "manual"
    UNIFORM = "uniform"
    PARAMETERS = "parameters"
    AUTO = "auto"


class PipelineScheduleType(Enum):
    SIMPLE_1F1B = "1f1b"
    INTERLEAVED_1F1B = "interleaved_1f1b"
    ZERO_BUBBLE = "zero_bubble_v"
    GPIPE = "gpipe"


class TracerType(Enum):
    VESCALE_FX = "vescale_fx"
    HF_FX = "hf_fx"
    TORCH_FX = "torch_fx"
    MANUAL = "manual"


@dataclass
class PipelineP2PSpec:
    """Explicit p2p input description for non-adjacent stage inputs
    (parity: legacy/vescale/plan/spec.py:74)."""

    peer_stage_idx: int
    peer_output_idx: int = 0
