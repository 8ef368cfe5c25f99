"""Zero-bubble schedule: backward split into B (input-grad, on the
critical path) and W (weight-grad, bubble filler).

Parity: legacy/vescale/pipe/_schedules/zero_bubble_v.py:132-1170
(ScheduledNode W/B-split backward) — a ZB-H1-style schedule: the 1F1B
skeleton emits BWD_B where 1F1B runs full BWD, and the deferred BWD_W
instructions fill the flush bubble at the end.  Registered through the
instruction registry (instruction.py), the extension mechanism the
reference exposes for custom schedules.

TRUE W/B split (no second graph traversal): a ZB_INIT instruction routes
every nn.Linear in the stage through pipe/wgrad_store._ZBLinearFn, whose
backward computes only dX and DEFERS the dW/db GEMMs into a
WeightGradStore keyed by (chunk, microbatch); BWD_B is one ordinary
torch.autograd.backward (non-linear params get their grads here, Linear
weight GEMMs are skipped), and BWD_W pops and runs the deferred closures.
No retain_graph — the graph is freed at B time; the store holds only the
(x, gy) pairs the W GEMMs need.
"""
from __future__ import annotations

from typing import List

import torch

from .instruction import Instr, one_f_one_b_schedule, register_instruction
from .wgrad_store import WeightGradStore, zb_patch_linears


def zero_bubble_schedule(stage: int, n_stages: int, n_mb: int) -> List[Instr]:
    base = one_f_one_b_schedule(stage, n_stages, n_mb)
    out: List[Instr] = [Instr("ZB_INIT", 0)]
    for ins in base:
        if ins.kind == "BWD":
            out.append(Instr("BWD_B", ins.microbatch, ins.chunk))
        else:
            out.append(ins)
    for m in range(n_mb):
        out.append(Instr("BWD_W", m))
    return out


@register_instruction("ZB_INIT")
def _zb_init(engine, ins):
    if getattr(engine, "_wgrad_store", None) is None:
        engine._wgrad_store = WeightGradStore()
        for ck in range(engine.V):
            zb_patch_linears(engine.stage.chunks[ck], engine._wgrad_store)


@register_instruction("BWD_B")
def _bwd_b(engine, ins):
    ck, m = ins.chunk, ins.microbatch
    out = engine._outputs.pop((ck, m))
    is_last = engine._is_last_global(ck)
    store: WeightGradStore = engine._wgrad_store
    store.begin()
    try:
        if is_last:
            out.backward()
        else:
            g = engine._recv_grads.pop((ck, m))
            torch.autograd.backward(out, grad_tensors=g)
    finally:
        store.end((ck, m))
    # input grads (sent upstream via _pop_input_grad) accumulated on the
    # leaf received activations by backward(); Linear dW/db deferred.


@register_instruction("BWD_W")
def _bwd_w(engine, ins):
    ck, m = ins.chunk, ins.microbatch
    engine._wgrad_store.pop_run((ck, m))
