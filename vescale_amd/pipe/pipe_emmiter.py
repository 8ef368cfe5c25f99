"""ScheduleEngine — emits and executes per-stage instruction lists.

Parity: legacy/vescale/pipe/pipe_emmiter.py:43-356 (PipelineEmitter +
ScheduleEngine) + the registered instruction impls of
_schedules/pipedream_flush.py:137-1290 — one engine executing the typed
Instr stream over the p2p layer.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..ndtimeline import ndtimeit, ndtimeit_p2p, predefined as ndm
from ..plan import PipelineParallelPlan, PipelineScheduleType
from . import p2p_communication as p2p
from .instruction import (
    Instr,
    gpipe_schedule,
    interleaved_1f1b_schedule,
    one_f_one_b_schedule,
)
from .pipe_stage import PipeModule


class ScheduleEngine:
    def __init__(
        self,
        stage: PipeModule,
        plan: PipelineParallelPlan,
        *,
        stage_to_rank: Optional[Callable[[int], int]] = None,
        pg=None,
        loss_fn: Optional[Callable] = None,
        device: Optional[torch.device] = None,
    ):
        self.stage = stage
        self.plan = plan
        self.P = plan.num_stages
        self.V = plan.virtual_chunks
        self.s = stage.stage_id
        self.pg = pg
        self.loss_fn = loss_fn
        self.device = device or (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        self.stage_to_rank = stage_to_rank or (lambda st: st)
        self.prev_rank = self.stage_to_rank(self.s - 1) if self.s > 0 else None
        self.next_rank = self.stage_to_rank(self.s + 1) if self.s < self.P - 1 else None
        # interleaved: the chunk chain wraps P-1 -> 0
        self.wrap_prev = self.stage_to_rank(self.P - 1)
        self.wrap_next = self.stage_to_rank(0)
        # chunk placement topology: "loop" (interleaved wrap) or "v" (ZB-V:
        # chunk 0 descends ranks, chunk 1 ascends; zbv.py module docstring)
        self.topology = (
            "v"
            if plan.schedule_type == PipelineScheduleType.ZERO_BUBBLE and self.V == 2
            else "loop"
        )
        # ZB-V: one communicator per (direction, chunk) stream.  A rank pair
        # can carry a forward stream AND a backward-grad stream of identical
        # shapes concurrently; RCCL/gloo match p2p by posting order per
        # (src, dst, communicator) — no tags on NCCL — so sharing one group
        # can cross-match a forward payload into a grad buffer.  Per-stream
        # groups make every channel single-stream and order-safe.
        self._zbv_pgs = None
        if self.topology == "v" and dist.is_initialized():
            self._zbv_pgs = {k: dist.new_group() for k in ("f0", "f1", "b0", "b1")}

        # per-(chunk, mb) state
        self._inputs: Dict[Tuple[int, int], Optional[torch.Tensor]] = {}
        self._outputs: Dict[Tuple[int, int], torch.Tensor] = {}
        self._recv_grads: Dict[Tuple[int, int], torch.Tensor] = {}
        self._losses: List[torch.Tensor] = []

    # ------------------------------------------------------------------
    def build_schedule(self, n_mb: int) -> List[Instr]:
        sched = self._build_schedule_impl(n_mb)
        # VESCALE_DUMP_INSTRUCTION (reference env flag): print this
        # stage's instruction stream for schedule debugging
        import os

        if os.environ.get("VESCALE_DUMP_INSTRUCTION"):
            import sys

            print(f"[pipe stage {self.s}] schedule ({len(sched)} instrs):",
                  file=sys.stderr)
            for i, ins in enumerate(sched):
                print(f"  {i:3d}: {ins}", file=sys.stderr)
        return sched

    def _build_schedule_impl(self, n_mb: int) -> List[Instr]:
        st = self.plan.schedule_type
        if st == PipelineScheduleType.SIMPLE_1F1B:
            if self.V != 1:
                raise ValueError("1F1B requires virtual_chunks == 1")
            return one_f_one_b_schedule(self.s, self.P, n_mb)
        if st == PipelineScheduleType.GPIPE:
            return gpipe_schedule(self.s, self.P, n_mb)
        if st == PipelineScheduleType.INTERLEAVED_1F1B:
            return interleaved_1f1b_schedule(self.s, self.P, n_mb, self.V)
        if st == PipelineScheduleType.ZERO_BUBBLE:
            if self.V == 1:
                from .zero_bubble import zero_bubble_schedule

                return zero_bubble_schedule(self.s, self.P, n_mb)
            if self.V == 2:
                from . import zero_bubble  # noqa: F401 — registers BWD_B/W
                from .zbv import zbv_schedule

                return zbv_schedule(self.s, self.P, n_mb)
            raise ValueError(
                "zero-bubble supports virtual_chunks 1 (ZB-H1) or 2 (ZB-V)"
            )
        raise ValueError(st)

    # global endpoints depend on the chunk topology --------------------
    def _is_first_global(self, ck: int) -> bool:
        return self.s == 0 and ck == 0

    def _is_last_global(self, ck: int) -> bool:
        if self.topology == "v":
            return self.s == 0 and ck == 1  # the V ascends back to rank 0
        return self.s == self.P - 1 and ck == self.V - 1

    # peer ranks for a chunk's in/out edges -----------------------------
    # "local" = same-rank chunk handoff (ZB-V rank P-1), handled by the
    # executor without touching the p2p layer.
    def _in_peer(self, ck: int):
        if self.topology == "v":
            if ck == 0:
                return self.prev_rank if self.s > 0 else None
            return "local" if self.s == self.P - 1 else self.stage_to_rank(self.s + 1)
        if self.s == 0:
            return self.wrap_prev if ck > 0 else None
        return self.prev_rank

    def _out_peer(self, ck: int):
        if self.topology == "v":
            if ck == 0:
                return "local" if self.s == self.P - 1 else self.next_rank
            return self.stage_to_rank(self.s - 1) if self.s > 0 else None
        if self.s == self.P - 1:
            return self.wrap_next if ck < self.V - 1 else None
        return self.next_rank

    # ------------------------------------------------------------------
    @staticmethod
    def _strip_backward(sched: List[Instr]) -> List[Instr]:
        """Forward-only projection of a schedule (reference
        looping_bfs.py:666,982 forward_only): drop every backward
        instruction, reduce fused ops to their forward half.  The
        resulting per-rank streams stay post-order matched because every
        rank drops the same halves."""
        out: List[Instr] = []
        for ins in sched:
            if ins.kind in ("BWD", "BWD_B", "BWD_W", "SEND_BWD", "RECV_BWD"):
                continue
            if ins.kind == "SEND_FWD_RECV_BWD":
                out.append(Instr("SEND_FWD", ins.microbatch, ins.chunk))
            elif ins.kind == "SEND_BWD_RECV_FWD":
                out.append(Instr("RECV_FWD", ins.microbatch2, ins.chunk))
            else:
                out.append(ins)
        return out

    def execute(
        self,
        minibatch: Optional[Tuple[torch.Tensor, torch.Tensor]],
        n_microbatches: int,
        *,
        grad_scale: Optional[float] = None,
        forward_only: bool = False,
    ):
        """Run one forward_backward over the minibatch split into
        n_microbatches.  First stage consumes inputs; last stage consumes
        targets + computes loss via loss_fn(output, target) (mean over
        microbatches).

        forward_only=True (or calling under torch.no_grad(), reference
        looping_bfs.py:892) runs just the forward halves — the PP eval /
        validation loop."""
        forward_only = forward_only or not torch.is_grad_enabled()
        self._forward_only = forward_only
        xs = ys = None
        if minibatch is not None:
            x, y = minibatch
            xs = list(torch.chunk(x, n_microbatches)) if x is not None else None
            ys = list(torch.chunk(y, n_microbatches)) if y is not None else None
        self._losses = []
        self._inputs.clear()
        self._outputs.clear()
        self._recv_grads.clear()
        scale = grad_scale if grad_scale is not None else 1.0 / n_microbatches

        sched = self.build_schedule(n_microbatches)
        if forward_only:
            sched = self._strip_backward(sched)
        for ins in sched:
            kind, m, ck = ins.kind, ins.microbatch, ins.chunk
            if kind == "RECV_FWD":
                with ndtimeit_p2p(ndm.RECV_FORWARD, self._in_peer(ck)):
                    t = p2p.recv_forward(
                        self._in_peer(ck), self._pg_for(kind, ck), device=self.device
                    )
                self._inputs[(ck, m)] = t if forward_only else t.requires_grad_(True)
            elif kind == "FWD":
                with ndtimeit(ndm.FORWARD_COMPUTE):
                    if forward_only:
                        with torch.no_grad():
                            self._fwd(ck, m, xs, ys, scale)
                    else:
                        self._fwd(ck, m, xs, ys, scale)
            elif kind == "SEND_FWD":
                peer = self._out_peer(ck)
                if peer == "local":
                    # ZB-V rank P-1: chunk 0 output feeds own chunk 1
                    h = self._outputs[(ck, m)].detach()
                    self._inputs[(ck + 1, m)] = (
                        h if forward_only else h.requires_grad_(True)
                    )
                else:
                    p2p.send_forward(
                        self._outputs[(ck, m)].detach(), peer, self._pg_for(kind, ck)
                    )
                if forward_only and not self._is_last_global(ck):
                    # no BWD will pop it — free the activation now
                    self._outputs.pop((ck, m), None)
            elif kind == "SEND_FWD_RECV_BWD":
                g = p2p.send_forward_recv_backward(
                    self._outputs[(ck, m)].detach(), self._out_peer(ck), self.pg,
                    device=self.device,
                )
                self._recv_grads[(ck, ins.microbatch2)] = g
            elif kind == "RECV_BWD":
                with ndtimeit_p2p(ndm.RECV_BACKWARD, self._out_peer(ck)):
                    self._recv_grads[(ck, m)] = p2p.recv_backward(
                        self._out_peer(ck), self._pg_for(kind, ck), device=self.device
                    )
            elif kind == "BWD":
                with ndtimeit(ndm.BACKWARD_COMPUTE):
                    self._bwd(ck, m)
            elif kind == "SEND_BWD":
                g = self._pop_input_grad(ck, m)
                peer = self._in_peer(ck)
                if peer == "local":
                    # ZB-V rank P-1: chunk 1 input grad feeds own chunk 0
                    self._recv_grads[(ck - 1, m)] = g
                else:
                    p2p.send_backward(g, peer, self._pg_for(kind, ck))
            elif kind == "SEND_BWD_RECV_FWD":
                g = self._pop_input_grad(ck, m)
                t = p2p.send_backward_recv_forward(
                    g, self._in_peer(ck), self.pg, device=self.device
                )
                self._inputs[(ck, ins.microbatch2)] = t.requires_grad_(True)
            else:
                from .instruction import VESCALE_INSTRUCTION_REGISTRY

                VESCALE_INSTRUCTION_REGISTRY[kind](self, ins)

        p2p.drain_send_reqs()
        if self._losses:
            return torch.stack([l.detach() for l in self._losses]).sum()
        return None

    # ------------------------------------------------------------------
    def _fwd(self, ck, m, xs, ys, scale):
        is_first_global = self._is_first_global(ck)
        is_last_global = self._is_last_global(ck)
        if is_first_global:
            inp = xs[m].to(self.device)
            self._inputs[(ck, m)] = None  # no upstream grad
            out = self.stage(inp, chunk=ck)
        else:
            inp = self._inputs[(ck, m)]
            out = self.stage(inp, chunk=ck)
        if is_last_global:
            assert self.loss_fn is not None, "last stage needs loss_fn"
            tgt = ys[m].to(self.device) if ys is not None else None
            loss = self.loss_fn(out, tgt) * scale
            self._outputs[(ck, m)] = loss
            self._losses.append(loss / scale)
        else:
            self._outputs[(ck, m)] = out

    def _bwd(self, ck, m):
        out = self._outputs.pop((ck, m))
        is_last_global = self._is_last_global(ck)
        if is_last_global:
            out.backward()
        else:
            g = self._recv_grads.pop((ck, m))
            torch.autograd.backward(out, grad_tensors=g)

    def _pg_for(self, kind: str, ck: int):
        if self._zbv_pgs is None:
            return self.pg
        key = f"f{ck}" if kind in ("RECV_FWD", "SEND_FWD") else f"b{ck}"
        return self._zbv_pgs[key]

    def _pop_input_grad(self, ck, m):
        inp = self._inputs.pop((ck, m))
        assert inp is not None, "first stage has no upstream grad to send"
        g = inp.grad
        assert g is not None, f"no input grad for chunk {ck} mb {m}"
        return g


def validate_pipeline_schedule(plan: PipelineParallelPlan) -> None:
    """Validate schedule/virtual-chunk consistency up front (reference
    pipe_emmiter.py:345) — same checks the engine applies lazily."""
    st = plan.schedule_type
    if st == PipelineScheduleType.INTERLEAVED_1F1B:
        assert plan.virtual_chunks > 1, "interleaved 1F1B needs virtual_chunks > 1"
    elif st == PipelineScheduleType.SIMPLE_1F1B:
        assert plan.virtual_chunks == 1, "1F1B needs virtual_chunks == 1"
    elif st == PipelineScheduleType.ZERO_BUBBLE:
        assert plan.virtual_chunks in (1, 2), (
            "zero-bubble supports virtual_chunks 1 (ZB-H1) or 2 (ZB-V)"
        )
