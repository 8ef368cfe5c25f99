"""Per-host unix-socket span collector.

Parity: legacy/vescale/ndtimeline/sock_streamer.py:94 (NDtimelineStreamer:
one collector process per host receives spans from every local rank over
a unix socket and runs the handler chain — e.g. merging all ranks into a
single Chrome trace).  The in-process handler chain in api.py remains the
default; this streamer is for multi-process-per-host runs.
"""
from __future__ import annotations

import json
import os
import socket
import socketserver
import threading
from typing import Callable, List, Optional

from .timer import Span

DEFAULT_SOCK = "/tmp/vescale_amd_ndtimeline.sock"


class _Handler(socketserver.StreamRequestHandler):
    def handle(self):
        for line in self.rfile:
            try:
                rec = json.loads(line)
                span = Span(
                    rec["metric"], rec["start_us"], rec["dur_us"], rec["rank"],
                    rec.get("step"),
                )
            except Exception:
                continue
            self.server._streamer._consume([span])  # type: ignore[attr-defined]


class NDtimelineStreamer:
    """Host-level collector: start() once per host (e.g. from local rank 0),
    then attach a SockHandler on every rank's NDTimerManager."""

    def __init__(self, sock_path: str = DEFAULT_SOCK):
        self.sock_path = sock_path
        self.handlers: List[Callable[[List[Span]], None]] = []
        self._server: Optional[socketserver.ThreadingUnixStreamServer] = None
        self._thread: Optional[threading.Thread] = None

    def _consume(self, spans: List[Span]):
        for h in self.handlers:
            try:
                h(spans)
            except Exception:
                pass

    def start(self):
        if os.path.exists(self.sock_path):
            os.unlink(self.sock_path)
        self._server = socketserver.ThreadingUnixStreamServer(
            self.sock_path, _Handler
        )
        self._server._streamer = self  # type: ignore[attr-defined]
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        if self._server:
            self._server.shutdown()
            self._server.server_close()
        if os.path.exists(self.sock_path):
            os.unlink(self.sock_path)


class SockHandler:
    """NDTimerManager handler shipping spans to the host streamer."""

    def __init__(self, sock_path: str = DEFAULT_SOCK):
        self.sock_path = sock_path
        self._sock: Optional[socket.socket] = None

    def _ensure(self):
        if self._sock is None:
            s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            s.connect(self.sock_path)
            self._sock = s

    def __call__(self, spans: List[Span]):
        try:
            self._ensure()
            payload = "".join(
                json.dumps(
                    dict(metric=s.metric, start_us=s.start_us, dur_us=s.dur_us,
                         rank=s.rank, step=s.step)
                ) + "\n"
                for s in spans
            )
            self._sock.sendall(payload.encode())
        except OSError:
            self._sock = None

    def close(self):
        if self._sock:
            self._sock.close()
            self._sock = None
