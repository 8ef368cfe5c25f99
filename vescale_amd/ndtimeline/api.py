"""ndtimeline public API (parity: legacy/vescale/ndtimeline/api.py:72-400
init_ndtimers / flush / wait)."""
from __future__ import annotations

import atexit
from typing import List, Optional

import torch.distributed as dist

from .handlers import ChromeTraceHandler, LocalRawHandler, LoggingHandler
from .timer import GlobalReferenceTime, NDMetricLevel, NDTimerManager

_chrome: Optional[ChromeTraceHandler] = None


def init_ndtimers(
    *,
    level: NDMetricLevel = NDMetricLevel.INFO,
    chrome_trace_path: Optional[str] = "ndtimeline_trace.json",
    raw_path: Optional[str] = None,
    log: bool = False,
) -> NDTimerManager:
    global _chrome
    # VESCALE_NDTIMELINE_LOG_LEVEL (reference env): override the metric
    # level by name, e.g. DEBUG/INFO/WARNING
    import os

    env_level = os.environ.get("VESCALE_NDTIMELINE_LOG_LEVEL")
    if env_level:
        try:
            level = NDMetricLevel[env_level.upper()]
        except KeyError:
            pass
    GlobalReferenceTime.calibrate()
    mgr = NDTimerManager(level)
    if chrome_trace_path:
        rank = dist.get_rank() if dist.is_initialized() else 0
        path = chrome_trace_path.replace(".json", f".rank{rank}.json")
        _chrome = ChromeTraceHandler(path)
        mgr.handlers.append(_chrome)
    if raw_path:
        mgr.handlers.append(LocalRawHandler(raw_path))
    if log:
        mgr.handlers.append(LoggingHandler())
    NDTimerManager.activate(mgr)
    atexit.register(lambda: (mgr.shutdown(), _chrome.dump() if _chrome else None))
    return mgr


def flush(step: Optional[int] = None):
    mgr = NDTimerManager.current()
    if mgr is None:
        return
    if step is not None:
        mgr.step = step
    mgr.flush()


def wait():
    mgr = NDTimerManager.current()
    if mgr is None:
        return
    mgr.wait()
    if _chrome is not None:
        _chrome.dump()
