"""Distributed timeline timers — HIP events + a globally-calibrated clock.

Parity: legacy/vescale/ndtimeline/timer.py:48-747 (GlobalReferenceTime
calibrated via collective sync, DeviceTimer with pooled device-event
pairs, NDTimerManager with background flush, ndtimer/ndtimeit/
ndtimeit_p2p decorators).  MI355X: torch.cuda.Event IS a HIP event on
ROCm; the calibration allreduce rides RCCL/gloo.
"""
from __future__ import annotations

import functools
import queue
import threading
import time
from dataclasses import dataclass, field
from enum import IntEnum
from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist


class NDMetricLevel(IntEnum):
    """Parity: timer.py:154 — metric verbosity tiers."""

    FRAMEWORK_DEBUG = 2
    USER_DEBUG = 3
    INFO = 10
    DISABLED = 100


@dataclass
class Span:
    metric: str
    start_us: float   # in the GLOBAL reference clock
    dur_us: float
    rank: int
    step: Optional[int] = None
    extra: dict = field(default_factory=dict)


class GlobalReferenceTime:
    """Cross-rank clock: after calibrate(), local_us() timestamps from all
    ranks share one zero (reference timer.py:48 — device-event global
    clock calibrated via collective allreduce)."""

    offset_us: float = 0.0
    calibrated: bool = False

    @classmethod
    def calibrate(cls, pg=None, iters: int = 5):
        if not dist.is_initialized():
            cls.offset_us = 0.0
            cls.calibrated = True
            return
        # barrier-bracketed exchange: approximate common instant = the
        # return of the barrier; offset = my clock at that instant vs mean
        samples = []
        for _ in range(iters):
            dist.barrier(group=pg)
            t = torch.tensor([time.perf_counter_ns() / 1e3], dtype=torch.float64)
            mine = float(t.item())
            dist.all_reduce(t, group=pg)
            mean = float(t.item()) / dist.get_world_size(pg)
            samples.append(mine - mean)
        samples.sort()
        cls.offset_us = samples[len(samples) // 2]
        cls.calibrated = True

    @classmethod
    def local_us(cls) -> float:
        return time.perf_counter_ns() / 1e3 - cls.offset_us


class DeviceTimer:
    """Per-metric pool of device event pairs (reference timer.py:226).
    On GPU, elapsed time is measured with HIP events (async, no host
    sync until flush); on CPU, perf_counter."""

    def __init__(self, metric: str, level: NDMetricLevel = NDMetricLevel.INFO,
                 extra: Optional[dict] = None):
        self.metric = metric
        self.level = level
        self.extra = extra or {}
        self._use_cuda = torch.cuda.is_available()
        self._pool: List[Tuple] = []
        self._inflight: List[Tuple] = []
        self._cpu_start: Optional[float] = None
        self._wall_start: Optional[float] = None

    def start(self, stream=None):
        self._wall_start = GlobalReferenceTime.local_us()
        if self._use_cuda:
            if self._pool:
                ev0, ev1 = self._pool.pop()
            else:
                ev0 = torch.cuda.Event(enable_timing=True)
                ev1 = torch.cuda.Event(enable_timing=True)
            ev0.record(stream or torch.cuda.current_stream())
            self._cur = (ev0, ev1)
        else:
            self._cpu_start = time.perf_counter_ns() / 1e3

    def stop(self, stream=None, step: Optional[int] = None):
        if self._use_cuda:
            ev0, ev1 = self._cur
            ev1.record(stream or torch.cuda.current_stream())
            self._inflight.append((ev0, ev1, self._wall_start, step))
        else:
            dur = time.perf_counter_ns() / 1e3 - self._cpu_start
            self._inflight.append((None, None, self._wall_start, step, dur))

    def flush(self, rank: int) -> List[Span]:
        out = []
        for rec in self._inflight:
            if self._use_cuda:
                ev0, ev1, wall, step = rec
                ev1.synchronize()
                dur = ev0.elapsed_time(ev1) * 1e3  # ms -> us
                self._pool.append((ev0, ev1))
            else:
                _, _, wall, step, dur = rec
            out.append(Span(self.metric, wall, dur, rank, step, dict(self.extra)))
        self._inflight.clear()
        return out


class NDTimerManager:
    """Owns all timers; a background thread drains flushed spans into the
    handler chain (reference timer.py:410 + sock_streamer handler chain)."""

    _instance: Optional["NDTimerManager"] = None

    def __init__(self, level: NDMetricLevel = NDMetricLevel.INFO):
        self.level = level
        self.timers: Dict[str, DeviceTimer] = {}
        self.spans: List[Span] = []
        self.handlers: List[Callable[[List[Span]], None]] = []
        self._q: "queue.Queue[List[Span]]" = queue.Queue()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._worker, daemon=True)
        self._thread.start()
        self.enabled = True
        self.step = 0

    @classmethod
    def current(cls) -> Optional["NDTimerManager"]:
        return cls._instance

    @classmethod
    def activate(cls, mgr: "NDTimerManager"):
        cls._instance = mgr

    def timer(self, metric: str, *, key: Optional[str] = None,
              extra: Optional[dict] = None) -> DeviceTimer:
        k = key or metric
        t = self.timers.get(k)
        if t is None:
            t = DeviceTimer(metric, extra=extra)
            self.timers[k] = t
        return t

    def _worker(self):
        while not self._stop.is_set():
            try:
                spans = self._q.get(timeout=0.2)
            except queue.Empty:
                continue
            self.spans.extend(spans)
            for h in self.handlers:
                try:
                    h(spans)
                except Exception:
                    pass

    def flush(self):
        rank = dist.get_rank() if dist.is_initialized() else 0
        batch: List[Span] = []
        for t in self.timers.values():
            batch.extend(t.flush(rank))
        if batch:
            self._q.put(batch)

    def wait(self):
        self.flush()
        while not self._q.empty():
            time.sleep(0.01)

    def shutdown(self):
        self.wait()
        self._stop.set()
        self._thread.join(timeout=2)


# ----------------------------- decorators ---------------------------------
def ndtimer(metric: str):
    def deco(fn):
        @functools.wraps(fn)
        def wrapper(*a, **k):
            mgr = NDTimerManager.current()
            if mgr is None or not mgr.enabled:
                return fn(*a, **k)
            t = mgr.timer(metric)
            t.start()
            try:
                return fn(*a, **k)
            finally:
                t.stop(step=mgr.step)

        return wrapper

    return deco


class ndtimeit:
    """Context-manager form (reference timer.py:716)."""

    def __init__(self, metric: str):
        self.metric = metric

    def __enter__(self):
        mgr = NDTimerManager.current()
        self.t = mgr.timer(self.metric) if mgr and mgr.enabled else None
        if self.t:
            self.t.start()
        return self

    def __exit__(self, *exc):
        if self.t:
            mgr = NDTimerManager.current()
            self.t.stop(step=mgr.step if mgr else None)
        return False


class ndtimeit_stream(ndtimeit):
    """Stream-aware variant: records the start/stop HIP events on a GIVEN
    stream (e.g. the FSDP all-gather/reduce-scatter comm streams), so the
    span measures the collective's device-side occupancy on its own stream
    rather than issue-side time on the compute stream (closes the
    reference's #4 comm-stream-accessor patch at engine level)."""

    def __init__(self, metric: str, stream):
        super().__init__(metric)
        self.stream = stream

    def __enter__(self):
        mgr = NDTimerManager.current()
        self.t = mgr.timer(self.metric) if mgr and mgr.enabled else None
        if self.t:
            self.t.start(stream=self.stream)
        return self

    def __exit__(self, *exc):
        if self.t:
            mgr = NDTimerManager.current()
            self.t.stop(stream=self.stream, step=mgr.step if mgr else None)
        return False


class ndtimeit_p2p(ndtimeit):
    """P2P-op timing with per-peer spans (reference p2p ndtimeit_p2p
    decorators, p2p_communication.py:624-847): each (metric, peer) pair
    gets its own timer so concurrent streams to different peers never mix
    event pairs, and every span carries extra={"peer": N} for the
    timeline/topology view."""

    def __init__(self, metric: str, peer):
        super().__init__(metric)
        self.peer = peer

    def __enter__(self):
        mgr = NDTimerManager.current()
        self.t = None
        if mgr and mgr.enabled:
            self.t = mgr.timer(
                self.metric,
                key=f"{self.metric}|peer={self.peer}",
                extra={"peer": self.peer},
            )
            self.t.start()
        return self
