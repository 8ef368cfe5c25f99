"""Span handlers: Chrome trace writer, logging, raw dump.

Parity: legacy/vescale/ndtimeline/handlers/ (chrome_trace_event.py:24-276
LocalTimelineNDHandler, logging_handler, local_raw_handler) — the Chrome
trace merges per-rank spans on the calibrated global clock so one
timeline shows all ranks (load into chrome://tracing or Perfetto;
rocprof traces interleave cleanly since both are us-scale JSON)."""
from __future__ import annotations

import json
import logging
import os
from typing import Dict, List

from .timer import Span

logger = logging.getLogger(__name__)


class ChromeTraceHandler:
    def __init__(self, path: str):
        self.path = path
        self.events: List[dict] = []

    def __call__(self, spans: List[Span]):
        for s in spans:
            self.events.append(
                {
                    "name": s.metric,
                    "ph": "X",
                    "ts": s.start_us,
                    "dur": s.dur_us,
                    "pid": f"rank{s.rank}",
                    "tid": s.metric.split("-")[0],
                    "args": {"step": s.step, **(s.extra or {})},
                }
            )

    def dump(self):
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        with open(self.path, "w") as f:
            json.dump({"traceEvents": self.events, "displayTimeUnit": "ms"}, f)


class LoggingHandler:
    def __call__(self, spans: List[Span]):
        for s in spans:
            logger.info("[ndtimeline] %s rank%d %.1fus", s.metric, s.rank, s.dur_us)


class LocalRawHandler:
    def __init__(self, path: str):
        self.path = path

    def __call__(self, spans: List[Span]):
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        with open(self.path, "a") as f:
            for s in spans:
                f.write(
                    json.dumps(
                        dict(metric=s.metric, start_us=s.start_us, dur_us=s.dur_us,
                             rank=s.rank, step=s.step)
                    )
                    + "\n"
                )


class MetricSummaryHandler:
    """Streaming per-metric aggregation (reference handlers/local_timer
    aggregation): count / total / mean / p50 / p99 over span durations,
    split per peer when spans carry one."""

    def __init__(self):
        self.durs: Dict[str, List[float]] = {}

    def __call__(self, spans: List[Span]):
        for s in spans:
            key = s.metric
            peer = (s.extra or {}).get("peer")
            if peer is not None:
                key = f"{s.metric}|peer={peer}"
            self.durs.setdefault(key, []).append(s.dur_us)

    @staticmethod
    def _pct(xs: List[float], q: float) -> float:
        i = min(len(xs) - 1, max(0, int(round(q * (len(xs) - 1)))))
        return xs[i]

    def summary(self) -> Dict[str, dict]:
        out = {}
        for k, xs in self.durs.items():
            ys = sorted(xs)
            out[k] = {
                "count": len(ys),
                "total_us": sum(ys),
                "mean_us": sum(ys) / len(ys),
                "p50_us": self._pct(ys, 0.5),
                "p99_us": self._pct(ys, 0.99),
            }
        return out
