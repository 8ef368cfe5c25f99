"""Instruction VM for pipeline schedules.

Parity: legacy/vescale/pipe/_schedules/instruction_base.py:58-571
(register_instruction, CommPacket, StageDeps, PipelineSchema,
InstructionBuilder) — redesigned as a compact typed-instruction list +
registry executed by the ScheduleEngine.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Dict, List

VESCALE_INSTRUCTION_REGISTRY: Dict[str, Callable] = {}


def register_instruction(name: str):
    def deco(fn):
        VESCALE_INSTRUCTION_REGISTRY[name] = fn
        return fn

    return deco


@dataclass
class Instr:
    kind: str            # RECV_FWD / FWD / SEND_FWD / RECV_BWD / BWD / SEND_BWD
    microbatch: int      # primary microbatch (the send side for fused ops)
    chunk: int = 0       # virtual-pipeline chunk index
    microbatch2: int = -1  # recv side of fused SEND_*_RECV_* ops

    def __repr__(self):
        return f"{self.kind}(mb={self.microbatch},ck={self.chunk})"


def gpipe_schedule(stage: int, n_stages: int, n_mb: int) -> List[Instr]:
    """All forwards then all backwards (parity: GPipe mode)."""
    out: List[Instr] = []
    for m in range(n_mb):
        if stage > 0:
            out.append(Instr("RECV_FWD", m))
        out.append(Instr("FWD", m))
        if stage < n_stages - 1:
            out.append(Instr("SEND_FWD", m))
    for m in range(n_mb):
        if stage < n_stages - 1:
            out.append(Instr("RECV_BWD", m))
        out.append(Instr("BWD", m))
        if stage > 0:
            out.append(Instr("SEND_BWD", m))
    return out


def one_f_one_b_schedule(stage: int, n_stages: int, n_mb: int) -> List[Instr]:
    """PipeDream-flush 1F1B (parity: _schedules/pipedream_flush.py:653-1110),
    Megatron-structured with FUSED bidirectional p2p in the steady state
    (one batch_isend_irecv keeps both xGMI link directions busy and is
    deadlock-free by construction).

    warmup = min(n_stages-1-stage, n_mb) forwards, steady 1F1B, cooldown."""
    out: List[Instr] = []
    warmup = min(n_stages - 1 - stage, n_mb)
    steady = n_mb - warmup
    last = stage == n_stages - 1
    first = stage == 0
    for f in range(warmup):
        if not first:
            out.append(Instr("RECV_FWD", f))
        out.append(Instr("FWD", f))
        if not last:
            out.append(Instr("SEND_FWD", f))
    if steady > 0 and not first:
        out.append(Instr("RECV_FWD", warmup))
    f, b = warmup, 0
    for i in range(steady):
        out.append(Instr("FWD", f))
        if not last:
            out.append(Instr("SEND_FWD_RECV_BWD", f, 0, b))
        out.append(Instr("BWD", b))
        if not first:
            if i == steady - 1:
                out.append(Instr("SEND_BWD", b))
            else:
                out.append(Instr("SEND_BWD_RECV_FWD", b, 0, f + 1))
        f += 1
        b += 1
    for _ in range(warmup):
        if not last:
            out.append(Instr("RECV_BWD", b))
        out.append(Instr("BWD", b))
        if not first:
            out.append(Instr("SEND_BWD", b))
        b += 1
    return out


def interleaved_1f1b_schedule(
    stage: int, n_stages: int, n_mb: int, n_chunks: int
) -> List[Instr]:
    """Interleaved (virtual-pipeline) 1F1B (parity:
    _schedules/looping_bfs.py).  Rank `stage` hosts chunks
    stage, stage+n_stages, ... in a vpp loop.  Simplified all-forward /
    all-backward per chunk-group ordering with microbatch grouping of
    n_stages (correctness-first; the GPU steady-state overlap comes from
    the p2p layer's batched ops)."""
    out: List[Instr] = []
    # forward sweep: chunks in order, each over all microbatches
    for ck in range(n_chunks):
        for m in range(n_mb):
            first_global = ck == 0 and stage == 0
            if not first_global:
                out.append(Instr("RECV_FWD", m, ck))
            out.append(Instr("FWD", m, ck))
            last_global = ck == n_chunks - 1 and stage == n_stages - 1
            if not last_global:
                out.append(Instr("SEND_FWD", m, ck))
    # backward sweep
    for ck in range(n_chunks - 1, -1, -1):
        for m in range(n_mb):
            last_global = ck == n_chunks - 1 and stage == n_stages - 1
            if not last_global:
                out.append(Instr("RECV_BWD", m, ck))
            out.append(Instr("BWD", m, ck))
            first_global = ck == 0 and stage == 0
            if not first_global:
                out.append(Instr("SEND_BWD", m, ck))
    return out
