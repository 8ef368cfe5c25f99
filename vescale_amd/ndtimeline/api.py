"""ndtimeline public API (parity: legacy/vescale/ndtimeline/api.py:72-400
init_ndtimers / flush / wait)."""
from __future__ import annotations

import atexit
from typing import Optional

import torch.distributed as dist

from .handlers import (
    ChromeTraceHandler,
    LocalRawHandler,
    LoggingHandler,
    MetricSummaryHandler,
)
from .timer import GlobalReferenceTime, NDMetricLevel, NDTimerManager

_chrome: Optional[ChromeTraceHandler] = None


def init_ndtimers(
    *,
    level: NDMetricLevel = NDMetricLevel.INFO,
    chrome_trace_path: Optional[str] = "ndtimeline_trace.json",
    raw_path: Optional[str] = None,
    log: bool = False,
    summary: bool = False,
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
    if summary:
        mgr.summary_handler = MetricSummaryHandler()
        mgr.handlers.append(mgr.summary_handler)
    NDTimerManager.activate(mgr)
    atexit.register(lambda: (mgr.shutdown(), _chrome.dump() if _chrome else None))
    return mgr


def calculate_topo(mesh) -> dict:
    """Infer the parallel topology for the timeline view (reference
    api.py:359 _calculate_topo): per-rank mesh coordinates + dim names so
    the trace UI can group rank timelines by (pp, dp, tp, ...) role.
    Accepts our DeviceMesh (or anything with .mesh tensor + dim names)."""
    import torch

    m = mesh.mesh if hasattr(mesh, "mesh") else torch.as_tensor(mesh)
    names = list(getattr(mesh, "mesh_dim_names", None) or
                 [f"dim{i}" for i in range(m.ndim)])
    topo = {}
    flat = m.reshape(-1)
    import itertools

    for coords in itertools.product(*[range(x) for x in m.shape]):
        rank = int(m[coords])
        topo[rank] = {names[i]: coords[i] for i in range(m.ndim)}
    return {"dims": names, "shape": list(m.shape), "rank_coords": topo}


def attach_topology(mesh) -> None:
    """Label the Chrome trace's per-rank process rows with their mesh
    coordinates (metadata events, ph='M')."""
    t = calculate_topo(mesh)
    if _chrome is None:
        return
    for rank, coords in t["rank_coords"].items():
        label = "/".join(f"{k}{v}" for k, v in coords.items())
        _chrome.events.append(
            {
                "name": "process_name",
                "ph": "M",
                "pid": f"rank{rank}",
                "args": {"name": f"rank{rank} [{label}]"},
            }
        )


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
