from .pipe_stage import PipeModule, build_shared_module_group, construct_pipeline_stage, construct_stage_modules
from .pipe_emmiter import ScheduleEngine, validate_pipeline_schedule
from .pipe_parser import (
    hf_symbolic_trace,
    parse_huggingface_model,
    split_graph_by_parameters,
)
from .p2p_communication import (
    recv_backward,
    recv_forward,
    send_backward,
    send_forward,
)

__all__ = [
    "PipeModule",
    "construct_pipeline_stage",
    "construct_stage_modules",
    "build_shared_module_group",
    "validate_pipeline_schedule",
    "ScheduleEngine",
    "hf_symbolic_trace",
    "parse_huggingface_model",
    "split_graph_by_parameters",
    "send_forward",
    "recv_forward",
    "send_backward",
    "recv_backward",
]
