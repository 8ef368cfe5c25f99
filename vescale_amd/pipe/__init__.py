from .pipe_stage import PipeModule, construct_pipeline_stage
from .pipe_emmiter import ScheduleEngine
from .p2p_communication import (
    recv_backward,
    recv_forward,
    send_backward,
    send_forward,
)

__all__ = [
    "PipeModule",
    "construct_pipeline_stage",
    "ScheduleEngine",
    "send_forward",
    "recv_forward",
    "send_backward",
    "recv_backward",
]
