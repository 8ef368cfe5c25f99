"""Pipeline-parallel tests (CPU/gloo ws=4): loss/grad parity vs single
device (mirrors legacy/test/parallel/pipeline/e2e/test_pp_accuracy_alignment.py)
for 1F1B, GPipe, interleaved, and zero-bubble schedules."""
import json
import os
import tempfile

import pytest
import torch
import torch.nn as nn

from tests.common import spawn

from vescale_amd.plan import (
    PipelineParallelPlan,
    PipelineScheduleType,
    PipelineSplitMethodType,
)


def _make_modules(seed=17, n=8, d=16):
    torch.manual_seed(seed)
    return [nn.Sequential(nn.Linear(d, d), nn.Tanh()) for _ in range(n)]


def _loss_fn(out, tgt):
    return (out - tgt).pow(2).mean()


def _single_device_ref(n_mb=4, bs=8, d=16):
    mods = _make_modules()
    model = nn.Sequential(*mods)
    torch.manual_seed(23)
    x = torch.randn(bs, d)
    y = torch.randn(bs, d)
    total = 0.0
    for xm, ym in zip(torch.chunk(x, n_mb), torch.chunk(y, n_mb)):
        loss = _loss_fn(model(xm), ym) / n_mb
        loss.backward()
        total += float(loss) * n_mb
    grads = {k: p.grad.clone() for k, p in model.named_parameters()}
    return total, grads


def _t_pp(rank, ws, sched, virtual_chunks, out_path):
    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    n_mb, bs, d = 4, 8, 16
    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=ws,
        virtual_chunks=virtual_chunks,
        schedule_type=PipelineScheduleType(sched),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, rank)
    engine = PipeEngine(stage, plan, loss_fn=_loss_fn, device=torch.device("cpu"))
    torch.manual_seed(23)
    x = torch.randn(bs, d)
    y = torch.randn(bs, d)
    loss = engine.forward_backward((x, y), n_mb)
    ref_loss, ref_grads = _single_device_ref(n_mb, bs, d)
    if rank == ws - 1:
        assert loss is not None
        assert abs(float(loss) - ref_loss) < 1e-6, (float(loss), ref_loss)
    # grad parity for this stage's params: map chunk modules back to the
    # global module list by identity of shapes + values is fragile; instead
    # compare against a fresh single-device backward on the same module
    # objects (they ARE the same objects, split in place)
    for name, p in stage.named_parameters():
        assert p.grad is not None, name
    # compare numerically: rebuild reference with same seed, walk in order
    ref_mods = _make_modules()
    ref_model = nn.Sequential(*ref_mods)
    torch.manual_seed(23)
    xr = torch.randn(bs, d)
    yr = torch.randn(bs, d)
    for xm, ym in zip(torch.chunk(xr, n_mb), torch.chunk(yr, n_mb)):
        (_loss_fn(ref_model(xm), ym) / n_mb).backward()
    # figure out which global modules this stage holds: UNIFORM split of 8
    # equal modules over (ws*virtual_chunks) parts
    n_parts = ws * virtual_chunks
    per = 8 // n_parts
    for ck in range(virtual_chunks):
        part_idx = ck * ws + rank
        gmods = ref_mods[part_idx * per : (part_idx + 1) * per]
        stage_chunk = stage.chunks[ck]
        sp = list(stage_chunk.parameters())
        rp = [p for m in gmods for p in m.parameters()]
        assert len(sp) == len(rp)
        for a, b in zip(sp, rp):
            assert torch.allclose(a.grad, b.grad, atol=1e-6), (
                sched, ck, (a.grad - b.grad).abs().max(),
            )


@pytest.mark.parametrize("sched,vc", [("1f1b", 1), ("gpipe", 1), ("interleaved_1f1b", 2), ("zero_bubble_v", 1)])
def test_pp_accuracy_alignment(sched, vc):
    spawn(4, _t_pp, sched, vc, None)


def test_uniform_split_balance():
    from vescale_amd.pipe.pipe_stage import uniform_split

    mods = _make_modules(n=8)
    parts = uniform_split(mods, 4)
    assert [len(p) for p in parts] == [2, 2, 2, 2]


def test_fx_pipe_parser():
    """GRAPH_EAGER: fx-trace an MLP stack and split into stage graphs."""
    import torch.nn as nn
    from vescale_amd.pipe.pipe_parser import construct_pipeline_split_graph

    torch.manual_seed(0)
    model = nn.Sequential(*[nn.Linear(8, 8) for _ in range(8)])
    stages = construct_pipeline_split_graph(model, ["2", "4", "6"], leaf_classes=("Linear",))
    assert len(stages) == 4
    x = torch.randn(3, 8)
    ref = model(x)
    h = x
    for st in stages:
        h = st(h)
        if isinstance(h, tuple):
            h = h[0]
    assert torch.allclose(h, ref, atol=1e-6)


def _t_pp_dp_3d(rank, ws, sched="1f1b"):
    """PP=2 x DP=2 (+ DistributedOptimizer ZeRO-2): loss parity vs single
    device (reference 4D-alignment methodology, minus TP).  Run with
    sched="zero_bubble_v" to cover deferred W-phase grads flowing into the
    DDP main_grad buffer (regression: closures once wrote param.grad
    directly, silently dropping Linear weight grads from DP reduction)."""
    import torch.distributed as dist
    from vescale_amd.ddp import DistributedDataParallel as DDP
    from vescale_amd.dtensor import init_device_mesh
    from vescale_amd.engine import PipeEngine
    from vescale_amd.optim import DistributedOptimizer
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    n_mb, bs, d = 4, 8, 16
    mesh = init_device_mesh("cpu", (2, 2), mesh_dim_names=("PP", "DP"))
    pp_rank, dp_rank = mesh.get_coordinate()
    dp_group = mesh.get_group(1)

    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=2,
        schedule_type=PipelineScheduleType(sched),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, pp_rank)
    ddp = DDP(stage, dp_group, use_distributed_optimizer=True,
              overlap_grad_reduce=False)  # PP = many backwards per step
    opt = DistributedOptimizer(
        torch.optim.AdamW(stage.parameters(), lr=1e-2), [ddp], clip_grad=0.0
    )
    engine = PipeEngine(
        stage, plan, loss_fn=_loss_fn,
        stage_to_rank=lambda s: s * 2 + dp_rank,
        device=torch.device("cpu"),
    )

    # reference run: full batch, single device, 2 steps
    ref_mods = _make_modules()
    ref_model = nn.Sequential(*ref_mods)
    ropt = torch.optim.AdamW(ref_model.parameters(), lr=1e-2)
    torch.manual_seed(23)
    xs = [torch.randn(bs, d) for _ in range(2)]
    ys = [torch.randn(bs, d) for _ in range(2)]
    ref_losses = []
    for x, y in zip(xs, ys):
        ropt.zero_grad()
        tot = 0.0
        for xm, ym in zip(torch.chunk(x, n_mb), torch.chunk(y, n_mb)):
            l = _loss_fn(ref_model(xm), ym) / n_mb
            l.backward()
            tot += float(l) * n_mb
        ropt.step()
        ref_losses.append(tot / n_mb)

    losses = []
    for x, y in zip(xs, ys):
        xd = torch.chunk(x, 2)[dp_rank]
        yd = torch.chunk(y, 2)[dp_rank]
        ddp.zero_grad_buffer()
        loss = engine.forward_backward((xd, yd), n_mb)
        ddp.finish_grad_sync()
        opt.step()
        if pp_rank == 1:
            l = loss.detach().clone() / n_mb
            dist.all_reduce(l, group=dp_group)
            losses.append(float(l) / 2)
    if pp_rank == 1:
        for a, b in zip(losses, ref_losses):
            assert abs(a - b) < 1e-5, (losses, ref_losses)


def test_pp_dp_zero2_3d():
    spawn(4, _t_pp_dp_3d)


def test_pp_dp_zero_bubble_wgrads_reduced():
    spawn(4, _t_pp_dp_3d, "zero_bubble_v")


def _t_shared_params(rank, ws):
    """Tied-embedding sync across stages (reference sync_shared_params)."""
    import torch.distributed as dist
    from vescale_amd.pipe.pipe_stage import PipeModule, build_shared_module_group

    emb = nn.Linear(4, 4, bias=False)
    stage = PipeModule([emb], rank, ws)
    build_shared_module_group(
        stage, [["weight"]], {"weight": 0 if rank == 0 else 1} if False else {"weight": rank},
        rank,
    )
    # both stages "own" a copy -> group over both, grads averaged by allreduce
    groups = build_shared_module_group(
        stage, [["weight"]], {"weight": rank}, rank
    )
    # simulate distinct grads, then sync
    emb.weight.grad = torch.full((4, 4), float(rank + 1))
    # group contains only this stage (stage_of_fqn maps to own stage) ->
    # rebuild with a group spanning both stages
    stage.shared_param_groups = [
        {"pg": dist.group.WORLD, "param": emb.weight, "fqns": ["weight"]}
    ]
    stage.sync_shared_params()
    assert torch.allclose(emb.weight.grad, torch.full((4, 4), 3.0))


def test_pp_shared_params_sync():
    spawn(2, _t_shared_params)


def _wgrad_split_body():
    from vescale_amd.pipe.wgrad_store import WeightGradStore, zb_patch_linears

    torch.manual_seed(0)
    ref = nn.Sequential(nn.Linear(6, 5), nn.GELU(), nn.Linear(5, 4))
    zb = nn.Sequential(nn.Linear(6, 5), nn.GELU(), nn.Linear(5, 4))
    zb.load_state_dict(ref.state_dict())
    store = WeightGradStore()
    assert zb_patch_linears(zb, store) == 2
    x = torch.randn(3, 6, requires_grad=True)
    xr = x.detach().clone().requires_grad_()
    ref(xr).square().sum().backward()
    store.begin()
    zb(x).square().sum().backward()   # no retain_graph
    store.end((0, 0))
    assert zb[0].weight.grad is None and zb[2].weight.grad is None
    assert torch.allclose(x.grad, xr.grad, atol=1e-6)
    store.pop_run((0, 0))
    for a, b in zip(zb.parameters(), ref.parameters()):
        assert torch.allclose(a.grad, b.grad, atol=1e-6)
    # inactive store: patched layers compute weight grads inline
    zb.zero_grad()
    zb(x.detach()).sum().backward()
    assert zb[0].weight.grad is not None
    print("WB_OK")


def test_wgrad_store_true_split():
    """Zero-bubble W/B split is REAL: after a B-phase backward the Linear
    weight grads are absent (deferred closures queued), input grads are
    exact, and the W-phase produces grads identical to a plain backward —
    with NO retain_graph.  Subprocess-isolated: parent-side autograd
    backward warms engine threads that poison later fork+gloo tests
    (tests/README.md fork-safety rule)."""
    import os
    import subprocess
    import sys

    code = (
        "import sys; sys.path.insert(0, %r); "
        "from tests.test_pipe import _wgrad_split_body; _wgrad_split_body()"
    ) % os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run([sys.executable, "-c", code], capture_output=True,
                         text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    assert "WB_OK" in out.stdout


# ---------------------------------------------------------------------------
# ZB-V: zero-bubble over the V-shaped 2-chunk placement (vescale_amd/pipe/
# zbv.py; reference zero_bubble_v.py:132-1170)
# ---------------------------------------------------------------------------
def test_zbv_timetable_properties():
    from vescale_amd.pipe.zbv import build_zbv_timetable, F, B, W

    for P, n_mb in ((2, 3), (4, 4), (4, 8)):
        tt = build_zbv_timetable(P, n_mb)
        assert len(tt) == P
        for s, ops in enumerate(tt):
            # every op exactly once per (cat, chunk, mb)
            assert len(ops) == 6 * n_mb
            assert len(set(ops)) == 6 * n_mb
            # per-(cat, chunk) microbatch order is monotonic
            seen = {}
            for cat, ck, m in ops:
                last = seen.get((cat, ck), -1)
                assert m == last + 1, (s, cat, ck, m, last)
                seen[(cat, ck)] = m
            # W never precedes its B on the same rank
            pos = {op: i for i, op in enumerate(ops)}
            for m in range(n_mb):
                for ck in (0, 1):
                    assert pos[(W, ck, m)] > pos[(B, ck, m)]
        # the W phases are interleaved into the stream, not all trailing
        # (zero-bubble property: W fills gaps) — check on rank 0
        ops0 = tt[0]
        first_w = min(i for i, (c, _, _) in enumerate(ops0) if c == W)
        last_b = max(i for i, (c, _, _) in enumerate(ops0) if c == B)
        assert first_w < last_b, "no W/B interleaving: schedule is GPipe-like"


def _t_zbv(rank, ws, n_mb=4):
    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    bs, d = 8, 16
    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=ws,
        virtual_chunks=2,
        schedule_type=PipelineScheduleType.ZERO_BUBBLE,
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, rank)
    engine = PipeEngine(stage, plan, loss_fn=_loss_fn, device=torch.device("cpu"))
    torch.manual_seed(23)
    x = torch.randn(bs, d)
    y = torch.randn(bs, d)
    loss = engine.forward_backward((x, y), n_mb)
    ref_loss, _ = _single_device_ref(n_mb, bs, d)
    if rank == 0:
        # ZB-V: rank 0 hosts the LAST chunk, so the loss lands here
        assert loss is not None
        assert abs(float(loss) - ref_loss) < 1e-6, (float(loss), ref_loss)
    # grad parity under the V placement: rank s holds parts s and 2P-1-s
    ref_mods = _make_modules()
    ref_model = nn.Sequential(*ref_mods)
    torch.manual_seed(23)
    xr = torch.randn(bs, d)
    yr = torch.randn(bs, d)
    for xm, ym in zip(torch.chunk(xr, n_mb), torch.chunk(yr, n_mb)):
        (_loss_fn(ref_model(xm), ym) / n_mb).backward()
    n_parts = ws * 2
    per = 8 // n_parts
    for ck in range(2):
        part_idx = rank if ck == 0 else n_parts - 1 - rank
        gmods = ref_mods[part_idx * per : (part_idx + 1) * per]
        sp = list(stage.chunks[ck].parameters())
        rp = [p for m in gmods for p in m.parameters()]
        assert len(sp) == len(rp)
        for a, b in zip(sp, rp):
            assert a.grad is not None
            assert torch.allclose(a.grad, b.grad, atol=1e-6), (
                ck, (a.grad - b.grad).abs().max(),
            )


@pytest.mark.parametrize("ws,n_mb", [(2, 4), (4, 4), (2, 1), (4, 2)])
def test_zbv_accuracy_alignment(ws, n_mb):
    # (2,1) and (4,2): fewer microbatches than stages — the bubbliest edge
    spawn(ws, _t_zbv, n_mb)


def _t_check_nan(rank, ws):
    """VESCALE_CHECK_NAN: a NaN injected into a stage's output is caught at
    the receiving boundary (reference p2p check_nan)."""
    import os

    import vescale_amd.pipe.p2p_communication as p2p

    os.environ["VESCALE_CHECK_NAN"] = "1"
    try:
        if rank == 0:
            t = torch.ones(4)
            t[2] = float("nan")
            p2p.send_forward(t, 1)
            p2p.drain_send_reqs()
        else:
            try:
                p2p.recv_forward(0)
            except FloatingPointError as e:
                assert "non-finite" in str(e)
            else:
                raise AssertionError("NaN not detected at the p2p boundary")
    finally:
        os.environ.pop("VESCALE_CHECK_NAN", None)


def test_p2p_check_nan():
    spawn(2, _t_check_nan)


# ---------------------------------------------------------------------------
# FULL 4D: PP2 x DP2 x TP2 (+ ZeRO-2) — the reference's flagship alignment
# methodology (legacy/examples/nanogpt_4D_finetune correctness story) at
# ws=8 on gloo: loss parity vs a single device over 2 optimizer steps.
# ---------------------------------------------------------------------------
class _ToLocal(nn.Module):
    """Boundary adapter: a TP-parallelized chunk's Replicate DTensor output
    becomes a plain tensor for the p2p layer (and plain inputs are lifted
    back by the dmodule hooks on the next stage)."""

    def __init__(self, inner):
        super().__init__()
        self.inner = inner

    def forward(self, x):
        out = self.inner(x)
        return out.to_local() if hasattr(out, "to_local") else out


def _t_pp_dp_tp_4d(rank, ws, sched="1f1b"):
    import torch.distributed as dist
    from vescale_amd.ddp import DistributedDataParallel as DDP
    from vescale_amd.dmodule import parallelize_module
    from vescale_amd.dtensor import Replicate, Shard, init_device_mesh
    from vescale_amd.engine import PipeEngine
    from vescale_amd.optim import DistributedOptimizer
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    n_mb, bs, d = 4, 8, 16
    mesh = init_device_mesh("cpu", (2, 2, 2), mesh_dim_names=("PP", "DP", "TP"))
    pp_rank, dp_rank, tp_rank = mesh.get_coordinate()
    dp_group = mesh.get_group(1)
    tp_mesh = mesh["TP"]

    # reference: full batch, single device
    ref_mods = _make_modules()
    ref_model = nn.Sequential(*ref_mods)
    ropt = torch.optim.AdamW(ref_model.parameters(), lr=1e-2)
    torch.manual_seed(23)
    xs = [torch.randn(bs, d) for _ in range(2)]
    ys = [torch.randn(bs, d) for _ in range(2)]
    ref_losses = []
    for x, y in zip(xs, ys):
        ropt.zero_grad()
        tot = 0.0
        for xm, ym in zip(torch.chunk(x, n_mb), torch.chunk(y, n_mb)):
            l = _loss_fn(ref_model(xm), ym) / n_mb
            l.backward()
            tot += float(l) * n_mb
        ropt.step()
        ref_losses.append(tot / n_mb)

    # 4D build: TP-parallelize every block, then PP-split, then DDP(+ZeRO2)
    mods = _make_modules()
    tp_plan = {
        "parameter": {
            r"0.weight": [Shard(0)],
            r"0.bias": [Shard(0)],
        },
        "forward": {
            "input": [[Replicate()]],
            r"1.output": [[Replicate()]],
        },
    }
    blocks = [_ToLocal(parallelize_module(m, tp_mesh, tp_plan)) for m in mods]
    plan = PipelineParallelPlan(
        num_stages=2,
        virtual_chunks=2 if sched == "zero_bubble_v" else 1,
        schedule_type=PipelineScheduleType(sched),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(blocks, plan, pp_rank)
    ddp = DDP(stage, dp_group, use_distributed_optimizer=True,
              overlap_grad_reduce=False)
    opt = DistributedOptimizer(
        torch.optim.AdamW(stage.parameters(), lr=1e-2), [ddp], clip_grad=0.0
    )
    engine = PipeEngine(
        stage, plan, loss_fn=_loss_fn,
        stage_to_rank=lambda s: s * 4 + dp_rank * 2 + tp_rank,
        device=torch.device("cpu"),
    )

    losses = []
    for x, y in zip(xs, ys):
        xd = torch.chunk(x, 2)[dp_rank]
        yd = torch.chunk(y, 2)[dp_rank]
        ddp.zero_grad_buffer()
        loss = engine.forward_backward((xd, yd), n_mb)
        for b in blocks:
            if hasattr(b.inner, "finish_grad_sync"):
                b.inner.finish_grad_sync()
        ddp.finish_grad_sync()
        opt.step()
        # loss lands on the last global chunk's rank: pp_rank P-1 for
        # 1F1B, pp_rank 0 for the V placement
        loss_pp = 0 if sched == "zero_bubble_v" else 1
        if pp_rank == loss_pp:
            l = loss.detach().clone() / n_mb
            dist.all_reduce(l, group=dp_group)
            losses.append(float(l) / 2)
    if pp_rank == (0 if sched == "zero_bubble_v" else 1):
        for a, b in zip(losses, ref_losses):
            assert abs(a - b) < 1e-4, (losses, ref_losses)


@pytest.mark.parametrize("sched", ["1f1b", "zero_bubble_v"])
def test_pp_dp_tp_4d(sched):
    spawn(8, _t_pp_dp_tp_4d, sched)


# ---------------------------------------------------------------------------
# user-defined schedules through the instruction registry (reference
# instruction/test_userdefine_schedule.py + test_pipe_instruction_register)
# ---------------------------------------------------------------------------
def _t_user_schedule(rank, ws):
    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.instruction import (
        VESCALE_INSTRUCTION_REGISTRY,
        Instr,
        register_instruction,
    )
    from vescale_amd.pipe.pipe_emmiter import ScheduleEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    n_mb, bs, d = 2, 4, 16
    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=ws,
        schedule_type=PipelineScheduleType.GPIPE,
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, rank)

    hits = {"n": 0}

    @register_instruction("USER_MARK")
    def _user_mark(engine, ins):
        hits["n"] += 1

    # a custom schedule: GPipe with USER_MARK instructions interleaved
    orig = ScheduleEngine._build_schedule_impl

    def custom(self, n_microbatches):
        sched = orig(self, n_microbatches)
        out = []
        for ins in sched:
            out.append(ins)
            if ins.kind == "FWD":
                out.append(Instr("USER_MARK", ins.microbatch))
        return out

    ScheduleEngine._build_schedule_impl = custom
    try:
        engine = PipeEngine(stage, plan, loss_fn=_loss_fn, device=torch.device("cpu"))
        torch.manual_seed(23)
        x = torch.randn(bs, d)
        y = torch.randn(bs, d)
        engine.forward_backward((x, y), n_mb)
        assert hits["n"] == n_mb, hits  # one mark per FWD on this stage
    finally:
        ScheduleEngine._build_schedule_impl = orig
        VESCALE_INSTRUCTION_REGISTRY.pop("USER_MARK", None)


def test_user_defined_schedule():
    spawn(2, _t_user_schedule)


# ---------------------------------------------------------------------------
# direct p2p API (reference pipeline/backend/test_p2p_comm.py scenarios:
# handshake vs known-shape, fused bidirectional, dtype fidelity, shape cache)
# ---------------------------------------------------------------------------
def _t_p2p_direct_api(rank, ws):
    import vescale_amd.pipe.p2p_communication as p2p

    dev = torch.device("cpu")
    # 1) handshake path (no shape known at the receiver), non-fp32 dtype
    if rank == 0:
        t = torch.arange(12, dtype=torch.bfloat16).reshape(3, 4) + 1
        p2p.send_forward(t, 1)
        p2p.drain_send_reqs()
    else:
        r = p2p.recv_forward(0)
        assert r.shape == (3, 4) and r.dtype == torch.bfloat16
        assert torch.equal(r, torch.arange(12, dtype=torch.bfloat16).reshape(3, 4) + 1)

    torch.distributed.barrier()
    # 2) known-shape path (no handshake)
    if rank == 0:
        t = torch.full((2, 5), 7.0)
        p2p.send_forward(t, 1, handshake=False)
        p2p.drain_send_reqs()
    else:
        r = p2p.recv_forward(0, shape=(2, 5), dtype=torch.float32, device=dev)
        assert torch.equal(r, torch.full((2, 5), 7.0))

    torch.distributed.barrier()
    # 3) backward direction with handshake
    if rank == 1:
        g = torch.randn(4, 3, generator=torch.Generator().manual_seed(1))
        p2p.send_backward(g, 0)
        p2p.drain_send_reqs()
    else:
        g = p2p.recv_backward(1)
        want = torch.randn(4, 3, generator=torch.Generator().manual_seed(1))
        assert torch.equal(g, want)

    torch.distributed.barrier()
    # channels re-handshake from here: the reuse cache pins one shape per
    # (pg, peer, direction) channel (static-shape invariant), and scenario
    # 4 reuses channels from 1/3 with different shapes
    p2p.reset_shape_cache()
    # 4) fused bidirectional pair, as the 1F1B steady state posts them:
    # stage i runs send_forward_recv_backward while stage i+1 runs
    # send_backward_recv_forward — both sides fused, matching post order
    if rank == 0:
        t = torch.ones(2, 2)
        g = p2p.send_forward_recv_backward(t, 1)
        assert torch.equal(g, torch.full((2, 2), 5.0))
    else:
        t = p2p.send_backward_recv_forward(torch.full((2, 2), 5.0), 0)
        assert torch.equal(t, torch.ones(2, 2))
        p2p.drain_send_reqs()

    torch.distributed.barrier()
    # 5) shape-cache reuse: same shape resent with VESCALE_REUSE_COMM_SHAPE
    os.environ["VESCALE_REUSE_COMM_SHAPE"] = "1"
    p2p.reset_shape_cache()
    try:
        for step in range(3):
            if rank == 0:
                p2p.send_forward(torch.full((3, 4), float(step)), 1)
                p2p.drain_send_reqs()
            else:
                r = p2p.recv_forward(0)
                assert torch.equal(r, torch.full((3, 4), float(step)))
    finally:
        del os.environ["VESCALE_REUSE_COMM_SHAPE"]
        p2p.reset_shape_cache()


def test_p2p_direct_api():
    spawn(2, _t_p2p_direct_api)


# ---------------------------------------------------------------------------
# forward-only (eval) projection of the schedules
# ---------------------------------------------------------------------------
def _t_pp_forward_only(rank, ws, sched, virtual_chunks):
    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    n_mb, bs, d = 4, 8, 16
    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=ws,
        virtual_chunks=virtual_chunks,
        schedule_type=PipelineScheduleType(sched),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, rank)
    engine = PipeEngine(stage, plan, loss_fn=_loss_fn, device=torch.device("cpu"))
    torch.manual_seed(23)
    x = torch.randn(bs, d)
    y = torch.randn(bs, d)

    # explicit forward_only kwarg
    loss = engine.evaluate((x, y), n_mb)
    # under torch.no_grad() the engine must auto-switch too
    with torch.no_grad():
        loss2 = engine.forward_backward((x, y), n_mb)

    # reference eval loss on a single device
    ref_model = nn.Sequential(*_make_modules())
    torch.manual_seed(23)
    xr = torch.randn(bs, d)
    yr = torch.randn(bs, d)
    with torch.no_grad():
        ref = sum(
            float(_loss_fn(ref_model(xm), ym))
            for xm, ym in zip(torch.chunk(xr, n_mb), torch.chunk(yr, n_mb))
        )
    loss_holder = 0 if sched == "zero_bubble_v" and virtual_chunks == 2 else ws - 1
    if rank == loss_holder:
        assert loss is not None and loss2 is not None
        assert abs(float(loss) - ref) < 1e-5, (float(loss), ref)
        assert abs(float(loss2) - ref) < 1e-5
    # NO grads were produced anywhere
    for _, p in stage.named_parameters():
        assert p.grad is None


@pytest.mark.parametrize(
    "sched,vc",
    [("1f1b", 1), ("gpipe", 1), ("interleaved_1f1b", 2), ("zero_bubble_v", 2)],
)
def test_pp_forward_only(sched, vc):
    spawn(2, _t_pp_forward_only, sched, vc)


def _t_single_stage_engine(rank, ws):
    """Degenerate P=1 pipeline: the engine runs pure compute, no p2p
    (reference pipeline/api/test_pipe_single_stage_ops.py)."""
    from vescale_amd.engine import PipeEngine
    from vescale_amd.pipe.pipe_stage import construct_pipeline_stage

    torch.manual_seed(1)
    mods = _make_modules()
    plan = PipelineParallelPlan(
        num_stages=1,
        virtual_chunks=1,
        schedule_type=PipelineScheduleType("1f1b"),
        split_method=PipelineSplitMethodType.UNIFORM,
    )
    stage = construct_pipeline_stage(mods, plan, rank)
    eng = PipeEngine(stage, plan, loss_fn=_loss_fn, device=torch.device("cpu"))
    torch.manual_seed(23)
    x = torch.randn(8, 16)
    y = torch.randn(8, 16)
    loss = eng.forward_backward((x, y), 4)
    # equals a plain n_mb-chunked loop on the same module stack
    ref_model = nn.Sequential(*_make_modules())
    torch.manual_seed(23)
    xr = torch.randn(8, 16)
    yr = torch.randn(8, 16)
    ref = sum(float(_loss_fn(ref_model(xm), ym))
              for xm, ym in zip(torch.chunk(xr, 4), torch.chunk(yr, 4)))
    assert abs(float(loss) - ref) < 1e-5
    for _, p in stage.named_parameters():
        assert p.grad is not None
    assert abs(float(eng.evaluate((x, y), 4)) - ref) < 1e-5


def test_single_stage_engine():
    spawn(1, _t_single_stage_engine)
