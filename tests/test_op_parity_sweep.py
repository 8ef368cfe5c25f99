"""Table-driven DTensor op parity sweep (ws=2, gloo): every op runs on
{Shard(0), Shard(1), Replicate} inputs and must reproduce the
single-device result after full_tensor().

Condenses the reference's per-op test files (legacy/test/dtensor/ops/
test_{pointwise,math,tensor,view,matrix}_ops.py) into one sweep; ops
with placement-specific semantics (reductions over the sharded dim,
softmax, matmul families, views) are the interesting rows.
"""
import pytest
import torch
import torch.nn.functional as F

from tests.common import spawn

from vescale_amd import Replicate, Shard, distribute_tensor, init_device_mesh

# (name, fn taking one [6, 8] tensor)
UNARY = [
    ("relu", torch.relu),
    ("sigmoid", torch.sigmoid),
    ("tanh", torch.tanh),
    ("exp", torch.exp),
    ("sqrt_abs", lambda t: torch.sqrt(t.abs() + 0.1)),
    ("clamp", lambda t: t.clamp(-0.5, 0.5)),
    ("softmax_last", lambda t: F.softmax(t, dim=-1)),
    ("log_softmax0", lambda t: F.log_softmax(t, dim=0)),
    ("sum_all", lambda t: t.sum()),
    ("sum_d0", lambda t: t.sum(0)),
    ("sum_d1_keep", lambda t: t.sum(1, keepdim=True)),
    ("mean_d1", lambda t: t.mean(1)),
    ("amax_d0", lambda t: t.amax(0)),
    ("argmax_d1", lambda t: t.argmax(1)),
    ("cumsum_d1", lambda t: t.cumsum(1)),
    ("transpose", lambda t: t.transpose(0, 1).contiguous()),
    ("reshape", lambda t: t.reshape(3, 16)),
    ("flatten", lambda t: t.flatten()),
    ("unsqueeze", lambda t: t.unsqueeze(1)),
    ("narrow", lambda t: t.narrow(1, 2, 4)),
    ("chunk0", lambda t: torch.chunk(t, 2, dim=0)[0]),
    ("split_cat", lambda t: torch.cat(torch.split(t, 4, dim=1), dim=1)),
    ("tril", lambda t: t[:6, :6].tril()),
    ("pow", lambda t: t.pow(2)),
    ("abs_neg", lambda t: (-t).abs()),
    ("var_d1", lambda t: t.var(1)),
    ("norm", lambda t: torch.linalg.vector_norm(t)),
    ("sort_d1", lambda t: torch.sort(t, dim=1)[0]),
    ("topk_d1", lambda t: torch.topk(t, 3, dim=1)[0]),
    ("max_d0", lambda t: t.max(0)[0]),
    ("stack_self", lambda t: torch.stack([t, t + 1], dim=0)),
    ("where_self", lambda t: torch.where(t > 0, t, torch.zeros(()))),
    ("type_f64", lambda t: t.double()),
    ("gelu", F.gelu),
    ("layer_norm", lambda t: F.layer_norm(t, (8,))),
    ("logaddexp_shift", lambda t: torch.logaddexp(t, t - 1)),
    ("masked_fill", lambda t: t.masked_fill(t > 0.5, 0.0)),
    ("repeat_interleave_d1", lambda t: t.repeat_interleave(2, dim=1)),
    ("flip_d1", lambda t: t.flip(1)),
    ("roll_d1", lambda t: t.roll(3, dims=1)),
    ("cummax_d1", lambda t: t.cummax(1)[0]),
    ("logsumexp_d1", lambda t: t.logsumexp(1)),
    ("clip_grad_style_sq", lambda t: (t * t).sum(-1, keepdim=True).sqrt()),
    ("erf", torch.erf),
    ("sinh_cosh", lambda t: t.sinh() + t.cosh()),
]

BINARY = [
    ("add", lambda a, b: a + b),
    ("mul", lambda a, b: a * b),
    ("sub_bcast_row", lambda a, b: a - b[0:1]),
    ("div_bcast_col", lambda a, b: a / (b[:, 0:1].abs() + 1)),
    ("matmul_t", lambda a, b: a @ b.t()),
    ("mm", lambda a, b: a @ b.reshape(8, 6)),
    ("maximum", torch.maximum),
    ("eq", lambda a, b: (a > b).float()),
]


def _t_sweep(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(11)
    x = torch.randn(6, 8)
    y = torch.randn(6, 8)
    placements = ([Shard(0)], [Shard(1)], [Replicate()])
    fails = []
    for name, fn in UNARY:
        ref = fn(x)
        for pl in placements:
            d = distribute_tensor(x, mesh, pl)
            try:
                out = fn(d)
                outs = out if isinstance(out, (tuple, list)) else (out,)
                refs = ref if isinstance(ref, (tuple, list)) else (ref,)
                for o, r in zip(outs, refs):
                    full = o.full_tensor() if hasattr(o, "full_tensor") else o
                    if not torch.allclose(full, r, atol=1e-5, equal_nan=True):
                        fails.append((name, pl, "value"))
            except Exception as e:
                fails.append((name, pl, f"{type(e).__name__}: {e}"))
    for name, fn in BINARY:
        ref = fn(x, y)
        for pa in placements:
            for pb in placements:
                da = distribute_tensor(x, mesh, pa)
                db = distribute_tensor(y, mesh, pb)
                try:
                    out = fn(da, db)
                    full = out.full_tensor() if hasattr(out, "full_tensor") else out
                    if not torch.allclose(full, ref, atol=1e-5):
                        fails.append((name, (pa, pb), "value"))
                except Exception as e:
                    fails.append((name, (pa, pb), f"{type(e).__name__}: {e}"))
    assert not fails, f"{len(fails)} failures:\n" + "\n".join(map(str, fails[:20]))


def test_op_parity_sweep():
    spawn(2, _t_sweep)


GRAD_OPS = [
    ("mul_sum", lambda t: (t * 3).sum()),
    ("relu_sum", lambda t: t.relu().sum()),
    ("softmax_ce", lambda t: F.log_softmax(t, -1)[..., 0].sum()),
    ("matmul", lambda t: (t @ t.t()).sum()),
    ("mean", lambda t: t.mean()),
    ("norm", lambda t: torch.linalg.vector_norm(t)),
    ("reshape_tanh", lambda t: t.reshape(-1).tanh().sum()),
    ("layer_norm", lambda t: F.layer_norm(t, (8,)).pow(2).sum()),
    ("sliced", lambda t: t[:, 2:6].sum()),
    ("gelu", lambda t: F.gelu(t).sum()),
    ("logsumexp", lambda t: t.logsumexp(1).sum()),
    ("masked_fill", lambda t: t.masked_fill(t > 0.5, 0.0).pow(2).sum()),
    ("cumsum_tanh", lambda t: t.cumsum(1).tanh().sum()),
    ("flip_mul", lambda t: (t.flip(1) * t).sum()),
    ("softplus", lambda t: F.softplus(t).sum()),
    ("var", lambda t: t.var(1).sum()),
]


def _t_grad_sweep(rank, ws):
    """Backward parity: d(loss)/d(input) through sharded ops must match the
    single-device gradient after full_tensor (the class of silent bug the
    4D test caught at the to_local boundary, checked per-op here)."""
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(13)
    x = torch.randn(6, 8)
    fails = []
    for name, fn in GRAD_OPS:
        xr = x.clone().requires_grad_(True)
        fn(xr).backward()
        for pl in ([Shard(0)], [Shard(1)], [Replicate()]):
            d = distribute_tensor(x, mesh, pl)
            d.requires_grad_(True)
            try:
                out = fn(d)
                loss = out.full_tensor() if hasattr(out, "full_tensor") else out
                loss.backward()
                g = d.grad
                gf = g.full_tensor() if hasattr(g, "full_tensor") else g
                if not torch.allclose(gf, xr.grad, atol=1e-5):
                    fails.append((name, pl, "grad value"))
            except Exception as e:
                fails.append((name, pl, f"{type(e).__name__}: {str(e)[:120]}"))
    assert not fails, f"{len(fails)} failures:\n" + "\n".join(map(str, fails[:20]))


def test_grad_parity_sweep():
    spawn(2, _t_grad_sweep)


def _t_ragged_interleaved_sweep(rank, ws):
    """Elementwise/norm ops on RaggedShard and InterleavedShard inputs
    (reference test/dtensor/ragged_shard/test_{elementwise,norm}.py and
    legacy shard/test_interleaved_shard.py)."""
    from vescale_amd import InterleavedShard, RaggedShard

    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(17)
    x = torch.randn(8, 6)
    cases = [
        [RaggedShard((0,), (1, 1))],
        [RaggedShard((0, 1), (1, 3))],   # uneven flat units
        [InterleavedShard(0, 2)],
    ]
    fns = [
        ("relu", torch.relu),
        ("scale_add", lambda t: t * 2 + 1),
        ("sum", lambda t: t.sum()),
        ("vec_norm", lambda t: torch.linalg.vector_norm(t)),
        ("sq_mean", lambda t: t.pow(2).mean()),
    ]
    fails = []
    for pl in cases:
        d = distribute_tensor(x, mesh, pl)
        for name, fn in fns:
            try:
                out = fn(d)
                full = out.full_tensor() if hasattr(out, "full_tensor") else out
                ref = fn(x)
                if not torch.allclose(full, ref, atol=1e-5):
                    fails.append((name, pl, "value"))
            except Exception as e:
                fails.append((name, pl, f"{type(e).__name__}: {str(e)[:120]}"))
        # round trip
        try:
            if not torch.allclose(d.full_tensor(), x):
                fails.append(("full_tensor", pl, "value"))
        except Exception as e:
            fails.append(("full_tensor", pl, f"{type(e).__name__}"))
    assert not fails, fails


def test_ragged_interleaved_sweep():
    spawn(2, _t_ragged_interleaved_sweep)


def _t_2d_mesh_sweep(rank, ws):
    """2D (2x2) mesh: ops with placements across BOTH mesh dims, catching
    cross-dim rule bugs the 1D sweeps cannot."""
    mesh = init_device_mesh("cpu", (2, 2))
    torch.manual_seed(19)
    x = torch.randn(8, 8)
    pls = [
        [Shard(0), Shard(1)],
        [Shard(1), Shard(0)],
        [Shard(0), Replicate()],
        [Replicate(), Shard(1)],
        [Shard(0), Shard(0)],   # chunk-of-chunk on one dim
    ]
    fns = [
        ("relu", torch.relu),
        ("softmax", lambda t: F.softmax(t, -1)),
        ("sum_all", lambda t: t.sum()),
        ("sum_d0", lambda t: t.sum(0)),
        ("mean_all", lambda t: t.mean()),
        ("transpose", lambda t: t.transpose(0, 1).contiguous()),
        ("matmul_self", lambda t: t @ t.t()),
        ("scale", lambda t: t * 0.5 + 2),
    ]
    fails = []
    for pl in pls:
        d = distribute_tensor(x, mesh, pl)
        if not torch.allclose(d.full_tensor(), x):
            fails.append(("full_tensor", pl, "roundtrip"))
            continue
        for name, fn in fns:
            try:
                out = fn(d)
                full = out.full_tensor() if hasattr(out, "full_tensor") else out
                if not torch.allclose(full, fn(x), atol=1e-5):
                    fails.append((name, pl, "value"))
            except Exception as e:
                fails.append((name, pl, f"{type(e).__name__}: {str(e)[:120]}"))
    assert not fails, f"{len(fails)}:\n" + "\n".join(map(str, fails[:15]))


def test_2d_mesh_sweep():
    spawn(4, _t_2d_mesh_sweep)


VIEW_FNS = [
    ("permute", lambda t: t.permute(1, 0).contiguous()),
    ("movedim", lambda t: t.movedim(0, 1).contiguous()),
    ("select0", lambda t: t.select(0, 3)),
    ("select1", lambda t: t.select(1, 5)),
    ("squeeze_chain", lambda t: t.unsqueeze(0).squeeze(0)),
    ("view_merge", lambda t: t.reshape(48)),
    ("view_split", lambda t: t.reshape(6, 2, 4)),
    ("view_back", lambda t: t.reshape(2, 3, 8).reshape(6, 8)),
    ("expand_bcast", lambda t: (t.unsqueeze(0).expand(3, 6, 8) + 0).sum(0)),
    ("cat_d0", lambda t: torch.cat([t, t], 0)),
    ("cat_d1", lambda t: torch.cat([t, t], 1)),
    ("slice_step", lambda t: t[::2]),
    ("flip", lambda t: t.flip(1)),
    ("roll", lambda t: t.roll(2, dims=1)),
    ("repeat_interleave", lambda t: t.repeat_interleave(2, dim=1)),
]


def _t_view_sweep(rank, ws):
    mesh = init_device_mesh("cpu", (ws,))
    torch.manual_seed(23)
    x = torch.randn(6, 8)
    fails = []
    for name, fn in VIEW_FNS:
        ref = fn(x)
        for pl in ([Shard(0)], [Shard(1)], [Replicate()]):
            d = distribute_tensor(x, mesh, pl)
            try:
                out = fn(d)
                full = out.full_tensor() if hasattr(out, "full_tensor") else out
                if not torch.allclose(full, ref, atol=1e-5):
                    fails.append((name, pl, "value"))
            except Exception as e:
                fails.append((name, pl, f"{type(e).__name__}: {str(e)[:120]}"))
    assert not fails, f"{len(fails)}:\n" + "\n".join(map(str, fails[:15]))


def test_view_sweep():
    spawn(2, _t_view_sweep)
