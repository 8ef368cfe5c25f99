"""DebugLogger — per-rank op/communication logging.

Parity: legacy/vescale/debug/debug_log.py:40-141 (VESCALE_DEBUG_MODE env,
per-rank file streams, log_communication + log_op).  The MI355X build
names it the same env for drop-in workflow parity:
  VESCALE_AMD_DEBUG_MODE=1          -> stderr
  VESCALE_AMD_DEBUG_MODE=/some/dir  -> per-rank files
"""
from __future__ import annotations

import os
import sys
from typing import Optional

import torch.distributed as dist


class DebugLogger:
    _enabled: Optional[bool] = None
    _stream = None
    _ranks = None

    @classmethod
    def enabled(cls) -> bool:
        if cls._enabled is None:
            mode = os.environ.get("VESCALE_AMD_DEBUG_MODE", os.environ.get("VESCALE_DEBUG_MODE", "0"))
            cls._enabled = mode not in ("0", "", "false", "False")
            if cls._enabled:
                rank = dist.get_rank() if dist.is_initialized() else 0
                if mode not in ("1", "true", "True"):
                    os.makedirs(mode, exist_ok=True)
                    cls._stream = open(os.path.join(mode, f"rank{rank}.log"), "a")
                else:
                    cls._stream = sys.stderr
        return cls._enabled

    @classmethod
    def set_vescale_debug_mode(cls, on: bool = True, rank_to_print=None):
        cls._enabled = on
        cls._ranks = rank_to_print
        if on and cls._stream is None:
            cls._stream = sys.stderr

    @classmethod
    def _emit(cls, msg: str):
        rank = dist.get_rank() if dist.is_initialized() else 0
        if cls._ranks is not None and rank not in cls._ranks:
            return
        print(f"[rank{rank}] {msg}", file=cls._stream or sys.stderr, flush=True)

    @classmethod
    def log_op(cls, op, args_info: str = ""):
        if cls.enabled():
            cls._emit(f"OP {op} {args_info}")

    @classmethod
    def log_communication(cls, kind: str, detail: str = ""):
        if cls.enabled():
            cls._emit(f"COMM {kind} {detail}")
