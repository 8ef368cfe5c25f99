"""Zero-bubble schedule: backward split into B (input-grad, on the
critical path) and W (weight-grad, bubble filler).

Parity: legacy/vescale/pipe/_schedules/zero_bubble_v.py:132-1170
(ScheduledNode W/B-split backward) — implemented as a ZB-H1-style
schedule: the 1F1B skeleton emits BWD_B where 1F1B runs full BWD, and
the deferred BWD_W instructions fill the flush bubble at the end.
Registered through the instruction registry (instruction.py), the
extension mechanism the reference exposes for custom schedules.
"""
from __future__ import annotations

from typing import List

import torch

from .instruction import Instr, one_f_one_b_schedule, register_instruction


def zero_bubble_schedule(stage: int, n_stages: int, n_mb: int) -> List[Instr]:
    base = one_f_one_b_schedule(stage, n_stages, n_mb)
    out: List[Instr] = []
    for ins in base:
        if ins.kind == "BWD":
            out.append(Instr("BWD_B", ins.microbatch, ins.chunk))
        else:
            out.append(ins)
    for m in range(n_mb):
        out.append(Instr("BWD_W", m))
    return out


@register_instruction("BWD_B")
def _bwd_b(engine, ins):
    ck, m = ins.chunk, ins.microbatch
    out = engine._outputs[(ck, m)]
    inp = engine._inputs.get((ck, m))
    is_last = engine.s == engine.P - 1 and ck == engine.V - 1
    g = None if is_last else engine._recv_grads.pop((ck, m))
    if inp is None:
        # first stage: no input grad needed; defer everything to W
        engine._w_state = getattr(engine, "_w_state", {})
        engine._w_state[(ck, m)] = (out, g)
        return
    gin = torch.autograd.grad(
        out, inp, grad_outputs=g, retain_graph=True, allow_unused=False
    )[0]
    inp.grad = gin
    engine._w_state = getattr(engine, "_w_state", {})
    engine._w_state[(ck, m)] = (out, g)


@register_instruction("BWD_W")
def _bwd_w(engine, ins):
    ck, m = ins.chunk, ins.microbatch
    out, g = engine._w_state.pop((ck, m))
    params = [p for p in engine.stage.chunks[ck].parameters() if p.requires_grad]
    if not params:
        return
    grads = torch.autograd.grad(out, params, grad_outputs=g, allow_unused=True)
    for p, gr in zip(params, grads):
        if gr is None:
            continue
        if p.grad is None:
            p.grad = gr
        else:
            p.grad = p.grad + gr
    engine._outputs.pop((ck, m), None)
