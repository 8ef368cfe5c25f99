"""ZB-V: zero-bubble schedule over the V-shaped 2-chunk placement.

Reference capability: legacy/vescale/pipe/_schedules/zero_bubble_v.py
(ScheduledNode V-schedule, CostGraph.try_v_schedule greedy builder,
:132-1170).  Re-designed here as a compact event-driven simulation:

Topology (V placement, virtual_chunks == 2):
  chunk 0 descends the ranks 0 -> P-1, chunk 1 ascends P-1 -> 0, so rank 0
  hosts both the model input (chunk 0) and the loss (chunk 1), and rank
  P-1 hands chunk-0 output to its own chunk 1 locally.  This is what kills
  the 1F1B warmup bubble: the rank that computes the loss is the rank that
  started, so B-phases become available everywhere almost immediately.

Schedule construction: a greedy global simulation with unit F/B/W costs.
At every step the earliest-startable op runs, ties broken B > F > W —
B drains the critical path, F feeds it, and W (the deferred Linear
weight-grad GEMMs, see wgrad_store.py) fills whatever idle time remains.
Because the engine's p2p sends are async (never block the instruction
stream) and recvs block, any dependency-valid global order executes
deadlock-free — so the simulated per-rank order IS the instruction list.

Dependencies simulated (cat in {F, B, W}):
  F(0,s,m): F(0,s-1,m)                       [s>0]
  F(1,s,m): F(1,s+1,m) | F(0,P-1,m) local    [s<P-1 | s==P-1]
  B(c,s,m): F(c,s,m) + grad producer:
    B(1,s,m): B(1,s-1,m) | loss=F(1,0,m)     [s>0 | s==0]
    B(0,s,m): B(0,s+1,m) | B(1,P-1,m) local  [s<P-1 | s==P-1]
  W(c,s,m): B(c,s,m)

Not modeled (honest deviations from the reference's CostGraph): comm
latency and the activation-memory bound.  Per-phase costs ARE a
parameter of build_zbv_timetable (feed measured means); the default unit
costs keep the builder deterministic and dependency-exact, and the
bubble structure (W fills idle) is preserved either way.  All ranks must
build with the SAME costs — the per-(direction, chunk) communicators
keep each channel single-stream so matching stays order-safe.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

from .instruction import Instr

F, B, W = 0, 1, 2


def build_zbv_timetable(
    P: int, n_mb: int, costs: Tuple[float, float, float] = (1.0, 1.0, 1.0)
) -> List[List[Tuple[int, int, int]]]:
    """Per-rank ordered op lists [(cat, chunk, mb), ...] for the V topology.

    costs = (f, b, w) phase durations — pass measured per-phase times (e.g.
    from ndtimeline's MetricSummaryHandler means) to tighten the simulated
    packing; the default unit costs preserve dependency order exactly and
    only affect which bubbles W phases land in."""
    end: Dict[Tuple[int, int, int, int], float] = {}
    order: List[List[Tuple[int, int, int]]] = [[] for _ in range(P)]
    cnt = [[0] * 6 for _ in range(P)]  # per-rank next mb, index cat*2+chunk
    t = [0.0] * P
    total = P * n_mb * 6

    def deps(cat, ck, s, m):
        d = []
        if cat == F:
            if ck == 0:
                if s > 0:
                    d.append((F, 0, s - 1, m))
            else:
                d.append((F, 1, s + 1, m) if s < P - 1 else (F, 0, P - 1, m))
        elif cat == B:
            d.append((F, ck, s, m))
            if ck == 1:
                if s > 0:
                    d.append((B, 1, s - 1, m))
            else:
                d.append((B, 0, s + 1, m) if s < P - 1 else (B, 1, P - 1, m))
        else:
            d.append((B, ck, s, m))
        return d

    scheduled = 0
    while scheduled < total:
        best = None
        for s in range(P):
            # candidate = next mb of each (cat, chunk), priority B > F > W
            for prio, (cat, ck) in enumerate(
                ((B, 0), (B, 1), (F, 0), (F, 1), (W, 0), (W, 1))
            ):
                m = cnt[s][cat * 2 + ck]
                if m >= n_mb:
                    continue
                ds = deps(cat, ck, s, m)
                if not all(x in end for x in ds):
                    continue
                start = max([t[s]] + [end[x] for x in ds])
                key = (start, prio, s)
                if best is None or key < best[0]:
                    best = (key, s, cat, ck, m)
        assert best is not None, "ZBV timetable: no ready op (cyclic deps?)"
        (start, _prio, _s), s, cat, ck, m = best
        fin = start + costs[cat]
        t[s] = fin
        end[(cat, ck, s, m)] = fin
        cnt[s][cat * 2 + ck] = m + 1
        order[s].append((cat, ck, m))
        scheduled += 1
    return order


def zbv_schedule(stage: int, P: int, n_mb: int) -> List[Instr]:
    """Instruction list for `stage` under ZB-V.  Comm placement: each recv
    sits immediately before its consuming compute, each send immediately
    after its producer; rank P-1's chunk0->chunk1 handoff and chunk1->chunk0
    grad handoff are local (the engine short-circuits peer == "local")."""
    order = build_zbv_timetable(P, n_mb)[stage]
    out: List[Instr] = [Instr("ZB_INIT", 0)]
    last = P - 1
    for cat, ck, m in order:
        if cat == F:
            first_global = stage == 0 and ck == 0
            local_in = stage == last and ck == 1
            if not first_global and not local_in:
                out.append(Instr("RECV_FWD", m, ck))
            out.append(Instr("FWD", m, ck))
            last_global = stage == 0 and ck == 1
            if not last_global:
                out.append(Instr("SEND_FWD", m, ck))
        elif cat == B:
            last_global = stage == 0 and ck == 1
            local_grad = stage == last and ck == 0
            if not last_global and not local_grad:
                out.append(Instr("RECV_BWD", m, ck))
            out.append(Instr("BWD_B", m, ck))
            first_global = stage == 0 and ck == 0
            if not first_global:
                out.append(Instr("SEND_BWD", m, ck))
        else:
            out.append(Instr("BWD_W", m, ck))
    return out
