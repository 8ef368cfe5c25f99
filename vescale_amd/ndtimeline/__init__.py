from .api import attach_topology, calculate_topo, flush, init_ndtimers, wait
from .timer import (
    DeviceTimer,
    GlobalReferenceTime,
    NDMetricLevel,
    NDTimerManager,
    ndtimeit,
    ndtimeit_stream,
    ndtimeit_p2p,
    ndtimer,
)
from . import predefined

__all__ = [
    "init_ndtimers",
    "calculate_topo",
    "attach_topology",
    "flush",
    "wait",
    "DeviceTimer",
    "NDTimerManager",
    "GlobalReferenceTime",
    "NDMetricLevel",
    "ndtimer",
    "ndtimeit",
    "ndtimeit_p2p",
    "predefined",
]
