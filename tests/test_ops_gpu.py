"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference
(SURVEY.md §4 — "numerics tests for a HIP kernel compare it against a plain
PyTorch fp32 reference of the same op")."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


def _ext():
    import vescale_amd.ops as ops

    return ops.require_ext()


@requires_gpu
def test_ext_loaded():
    import vescale_amd.ops as ops

    assert ops.has_ext(), f"HIP extension must be present on GPU: {ops._import_error}"


@requires_gpu
def test_rmsnorm_fwd_bwd():
    from vescale_amd.ops import rmsnorm

    torch.manual_seed(0)
    x = torch.randn(64, 4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = rmsnorm(x, w, 1e-5)
    dy = torch.randn_like(out)
    out.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    rrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
    ref = xf * rrms * wf
    ref.backward(dy.float())

    assert torch.allclose(out.float(), ref.detach(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(x.grad.float(), xf.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=1.0, rtol=2e-2)


@requires_gpu
def test_rope():
    from vescale_amd.ops import build_rope_table, rope_apply
    from vescale_amd.ops.functional import _rope_ref

    torch.manual_seed(1)
    tab = build_rope_table(128, 128, 500000.0, device="cuda")
    x = torch.randn(2, 128, 8, 128, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = rope_apply(x, tab)
    ref = _rope_ref(x.detach().float(), tab, 0, False)
    assert torch.allclose(out.float(), ref, atol=2e-2, rtol=2e-2)
    dy = torch.randn_like(out)
    out.backward(dy)
    ref_grad = _rope_ref(dy.float(), tab, 0, True)
    assert torch.allclose(x.grad.float(), ref_grad, atol=2e-2, rtol=2e-2)


@requires_gpu
def test_swiglu():
    from vescale_amd.ops import swiglu

    torch.manual_seed(2)
    g = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    u = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = swiglu(g, u)
    dy = torch.randn_like(out)
    out.backward(dy)
    gf = g.detach().float().requires_grad_(True)
    uf = u.detach().float().requires_grad_(True)
    ref = torch.nn.functional.silu(gf) * uf
    ref.backward(dy.float())
    assert torch.allclose(out.float(), ref.detach(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(g.grad.float(), gf.grad, atol=3e-2, rtol=3e-2)
    assert torch.allclose(u.grad.float(), uf.grad, atol=3e-2, rtol=3e-2)


@requires_gpu
def test_swiglu_packed():
    from vescale_amd.ops import swiglu_packed

    torch.manual_seed(21)
    gu = torch.randn(64, 2048, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = swiglu_packed(gu)
    dy = torch.randn_like(out)
    out.backward(dy)
    F = 1024
    gf = gu.detach().float().requires_grad_(True)
    ref = torch.nn.functional.silu(gf[..., :F]) * gf[..., F:]
    ref.backward(dy.float())
    assert torch.allclose(out.float(), ref.detach(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(gu.grad.float(), gf.grad, atol=3e-2, rtol=3e-2)


@requires_gpu
def test_adamw_clip_fused():
    from vescale_amd.ops import adamw_step_flat

    torch.manual_seed(22)
    n = 4096
    p = torch.randn(n, device="cuda").bfloat16()
    master = p.float().clone()
    g = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    clip = torch.tensor([0.5], device="cuda")
    adamw_step_flat(p, master, g, m, v, lr=1e-3, step=1, clip_scale=clip)
    # reference with pre-scaled grad
    p2 = torch.randn(0)
    m2 = torch.zeros(n, device="cuda")
    v2 = torch.zeros(n, device="cuda")
    gf = g.float() * 0.5
    m2.add_(gf, alpha=0.1)
    assert torch.allclose(m, m2, atol=1e-6)


@requires_gpu
def test_fused_cross_entropy():
    from vescale_amd.ops import fused_cross_entropy

    torch.manual_seed(3)
    N, V = 512, 128256
    logits = torch.randn(N, V, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    tgt = torch.randint(0, V, (N,), device="cuda")
    tgt[:7] = -100
    loss = fused_cross_entropy(logits, tgt)
    loss.backward()

    lf = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lf, tgt, ignore_index=-100)
    ref.backward()
    assert torch.allclose(loss.float(), ref.detach(), atol=1e-2, rtol=1e-3), (float(loss), float(ref))
    assert torch.allclose(logits.grad.float(), lf.grad, atol=1e-3, rtol=5e-2)


@requires_gpu
def test_adamw_flat():
    from vescale_amd.ops import adamw_step_flat

    torch.manual_seed(4)
    n = 4096 * 9 + 17
    p32 = torch.randn(n, device="cuda")
    p = p32.bfloat16().clone()
    master = p.float().clone()
    g = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    for step in range(1, 4):
        adamw_step_flat(p, master, g, m, v, lr=1e-3, beta1=0.9, beta2=0.95,
                        eps=1e-8, weight_decay=0.1, step=step)
    # fp32 torch reference
    pr = p32.bfloat16().float().clone()
    mr = torch.zeros(n, device="cuda")
    vr = torch.zeros(n, device="cuda")
    gf = g.float()
    for step in range(1, 4):
        pr.mul_(1 - 1e-3 * 0.1)
        mr.mul_(0.9).add_(gf, alpha=0.1)
        vr.mul_(0.95).addcmul_(gf, gf, value=0.05)
        bc1 = 1 - 0.9**step
        bc2 = 1 - 0.95**step
        pr.addcdiv_(mr, (vr / bc2).sqrt().add_(1e-8), value=-1e-3 / bc1)
    assert torch.allclose(master, pr, atol=1e-5, rtol=1e-4)
    assert torch.allclose(m, mr, atol=1e-6)
    assert torch.allclose(v, vr, atol=1e-7)


@requires_gpu
def test_l2norm_and_scale():
    from vescale_amd.ops import l2norm_sq, scale_flat_

    torch.manual_seed(5)
    x = torch.randn(123457, device="cuda", dtype=torch.bfloat16)
    got = l2norm_sq(x)
    ref = x.float().pow(2).sum()
    assert torch.allclose(got, ref, rtol=1e-3)
    y = x.clone()
    scale_flat_(y, 0.5)
    assert torch.allclose(y.float(), (x.float() * 0.5), atol=1e-2)


@requires_gpu
def test_philox_shard_parity():
    """The headline RNG property: a sharded fill is bitwise-identical to the
    single-GPU fill (reference patch #2 semantics)."""
    C = _ext()
    n = 1 << 16
    seed, off = 1234, 7
    full = torch.empty(n, device="cuda", dtype=torch.bfloat16)
    C.philox_uniform_(full, [n], [n], [0], 0, True, seed, off, 0.0, 1.0)
    # shard = second quarter, filled independently with flat offset
    q = n // 4
    shard = torch.empty(q, device="cuda", dtype=torch.bfloat16)
    C.philox_uniform_(shard, [n], [q], [q], q, True, seed, off, 0.0, 1.0)
    assert torch.equal(shard, full[q : 2 * q])
    # 2-D sharding: rows [8:16) of a [32, 64] matrix
    g = torch.empty(32 * 64, device="cuda", dtype=torch.bfloat16)
    C.philox_uniform_(g, [32 * 64], [32 * 64], [0], 0, True, seed, off, 0.0, 1.0)
    g = g.view(32, 64)
    sh = torch.empty(8, 64, device="cuda", dtype=torch.bfloat16)
    C.philox_uniform_(sh.view(-1), [32, 64], [8, 64], [8, 0], 0, False, seed, off, 0.0, 1.0)
    assert torch.equal(sh, g[8:16])
    # column shard: cols [16:32)
    sc = torch.empty(32, 16, device="cuda", dtype=torch.bfloat16)
    C.philox_uniform_(sc.view(-1), [32, 64], [32, 16], [0, 16], 0, False, seed, off, 0.0, 1.0)
    assert torch.equal(sc, g[:, 16:32])


@requires_gpu
def test_philox_normal_stats():
    C = _ext()
    n = 1 << 22
    out = torch.empty(n, device="cuda", dtype=torch.float32)
    C.philox_normal_(out, [n], [n], [0], 0, True, 99, 0, 0.0, 1.0)
    assert abs(out.mean().item()) < 5e-3
    assert abs(out.std().item() - 1.0) < 5e-3


@requires_gpu
def test_philox_dropout():
    C = _ext()
    torch.manual_seed(6)
    x = torch.ones(1 << 20, device="cuda", dtype=torch.bfloat16)
    out, mask = C.philox_dropout(x, [x.numel()], [x.numel()], [0], 0, True, 7, 0, 0.1, True)
    keep_frac = mask.float().mean().item()
    assert abs(keep_frac - 0.9) < 5e-3
    # kept elements scaled by 1/(1-p)
    kept = out[mask.bool()]
    assert torch.allclose(kept.float(), torch.full_like(kept.float(), 1.0 / 0.9), atol=1e-2)
    dropped = out[~mask.bool()]
    assert (dropped == 0).all()


@requires_gpu
def test_gemm_tn_mfma():
    """Hand-written MFMA GEMM numerics vs fp32 reference (all variants)."""
    C = _ext()
    torch.manual_seed(30)
    a = torch.randn(512, 1024, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(512, 1024, device="cuda", dtype=torch.bfloat16)
    ref = a.float() @ b.float().t()
    for v in (0, 2, 3):
        got = C.gemm_tn(a, b, v).float()
        assert torch.allclose(got, ref, atol=2.0, rtol=2e-2), f"variant {v}"
    # identity with asymmetric B (transpose-detecting, guide G9)
    eye = torch.zeros(256, 256, device="cuda", dtype=torch.bfloat16)
    eye[range(256), range(256)] = 1.0
    bb = (torch.arange(256, device="cuda").view(-1, 1) * 0.1
          + torch.arange(256, device="cuda").view(1, -1) * 0.001).bfloat16()
    got = C.gemm_tn(eye, bb, 0).float()
    assert torch.allclose(got, bb.float().t(), atol=0.05)


@requires_gpu
def test_fused_add_rmsnorm():
    """Fused residual-add + rmsnorm (fwd + dres-fused bwd) vs fp32 ref."""
    import vescale_amd.ops as ops
    from vescale_amd.ops.functional import fused_add_rmsnorm

    ops.require_ext()
    torch.manual_seed(5)
    x = torch.randn(64, 4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    r = torch.randn(64, 4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(4096, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y, res_new = fused_add_rmsnorm(x, r, w, 1e-5)
    dy = torch.randn_like(y)
    dr2 = torch.randn_like(y)
    (y * dy + res_new * dr2).sum().backward()

    xf = x.detach().float().requires_grad_()
    rf = r.detach().float().requires_grad_()
    wf = w.detach().float().requires_grad_()
    rn = xf + rf
    yf = rn * torch.rsqrt(rn.pow(2).mean(-1, keepdim=True) + 1e-5) * wf
    (yf * dy.float() + rn * dr2.float()).sum().backward()
    assert torch.allclose(y.float(), yf.float(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(res_new.float(), rn.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(x.grad.float(), xf.grad, atol=8e-2, rtol=8e-2)
    assert torch.allclose(r.grad.float(), rf.grad, atol=8e-2, rtol=8e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=2.0, rtol=5e-2)


@requires_gpu
def test_flash_attention_fwd_kernel():
    """fa_fwd MFMA forward kernel (attention_fwd.hip): O and logsumexp vs
    fp32 reference (causal, GQA, D=128). The kernel's layout assumptions
    are themselves probe-verified on HW by tools/fa_fwd_check.py /
    tools/tr_probe_check.py."""
    import math

    import torch.nn.functional as F

    import vescale_amd.ops as ops

    C = ops.require_ext()
    torch.manual_seed(11)
    B, Hq, Hkv, S, D = 2, 4, 2, 512, 128
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
    sc = 1.0 / math.sqrt(D)
    o, lse = C.fa_fwd(q, k, v, sc)
    kx = k.float().repeat_interleave(Hq // Hkv, 1)
    vx = v.float().repeat_interleave(Hq // Hkv, 1)
    ref = F.scaled_dot_product_attention(q.float(), kx, vx, is_causal=True)
    err = (o.float() - ref).abs() / ref.abs().clamp_min(1.0)
    assert err.max().item() < 2e-2, err.max().item()
    s_full = torch.einsum("bhqd,bhkd->bhqk", q.float(), kx) * sc
    mask = torch.full((S, S), float("-inf"), device="cuda").triu(1)
    want_lse = (s_full + mask).logsumexp(-1)
    assert (lse - want_lse).abs().max().item() < 1e-2


@requires_gpu
def test_flash_attention_train_path():
    """flash_attention_causal (our fwd kernel + library flash backward fed
    with our logsumexp) vs torch SDPA autograd (causal, GQA, D=128)."""
    import torch.nn.functional as F
    from vescale_amd.ops import flash_attention_causal

    torch.manual_seed(7)
    B, Hq, Hkv, S, D = 2, 8, 2, 512, 128
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    out = flash_attention_causal(q, k, v)
    out.backward(dy)
    gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
    q.grad = k.grad = v.grad = None
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
    ref.backward(dy)
    assert torch.allclose(out.float(), ref.float(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(gq.float(), q.grad.float(), atol=5e-2, rtol=8e-2)
    assert torch.allclose(gk.float(), k.grad.float(), atol=8e-2, rtol=1e-1)
    assert torch.allclose(gv.float(), v.grad.float(), atol=8e-2, rtol=1e-1)


@requires_gpu
def test_flash_attention_aten_fallback():
    """VESCALE_FA=aten must route flash_attention_causal to the library
    kernels and agree with the HIP path."""
    import importlib
    import os

    import vescale_amd.ops.functional as F_

    torch.manual_seed(4)
    q = torch.randn(1, 4, 256, 128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(1, 2, 256, 128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(1, 2, 256, 128, device="cuda", dtype=torch.bfloat16)
    hip = F_.flash_attention_causal(q, k, v)
    os.environ["VESCALE_FA"] = "aten"
    try:
        lib = F_.flash_attention_causal(q, k, v)
    finally:
        os.environ.pop("VESCALE_FA", None)
    assert torch.allclose(hip.float(), lib.float(), atol=3e-2, rtol=3e-2)


@requires_gpu
def test_gemm8_tn():
    """8-phase 256x256 MFMA GEMM (gemm8.hip) vs fp32 reference."""
    import vescale_amd.ops as ops

    C = ops.require_ext()
    torch.manual_seed(2)
    a = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16) / 8
    b = torch.randn(256, 256, device="cuda", dtype=torch.bfloat16) / 8
    c = C.gemm_tn8(a, b)
    ref = a.float() @ b.float().t()
    rel = ((c.float() - ref).abs() / ref.abs().clamp_min(1e-2)).max().item()
    assert rel < 0.05, rel


@requires_gpu
def test_flash_attention_bwd_kernel():
    """Direct test of our fa_bwd MFMA kernel (attention_bwd.hip) — kept as
    an alternative backward (slower than the library's today, see
    NOTES_ROUND2.md) — vs SDPA autograd grads."""
    import math

    import torch.nn.functional as F

    import vescale_amd.ops as ops

    C = ops.require_ext()
    torch.manual_seed(13)
    B, Hq, Hkv, S, D = 1, 4, 2, 256, 128
    q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    dy = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
    sc = 1.0 / math.sqrt(D)
    o, lse = C.fa_fwd(q.detach(), k.detach(), v.detach(), sc)
    dq, dk, dv = C.fa_bwd(q.detach(), k.detach(), v.detach(), o, dy, lse, sc)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
    ref.backward(dy)
    assert torch.allclose(dq.float(), q.grad.float(), atol=6e-2, rtol=8e-2)
    assert torch.allclose(dk.float(), k.grad.float(), atol=8e-2, rtol=1e-1)
    assert torch.allclose(dv.float(), v.grad.float(), atol=8e-2, rtol=1e-1)


@requires_gpu
def test_sharded_dropout_bitwise_tp_parity():
    """End-to-end headline RNG property through DTensor dispatch: dropout
    on a SHARDED DTensor is bitwise-identical to the same dropout on the
    full tensor (reference nanoGPT TP4 dropout parity)."""
    import os
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29621")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    from vescale_amd.dtensor import (
        DTensor,
        Replicate,
        Shard,
        distribute_tensor,
        init_device_mesh,
    )
    from vescale_amd.dtensor.random import init_rng_tracker

    mesh = init_device_mesh("cpu", (1,))  # ws=1: shard == slice of full
    tr = init_rng_tracker(mesh, "thread", seed=777)

    x = torch.ones(64, 128, device="cuda", dtype=torch.bfloat16)
    # full-tensor dropout
    d_full = distribute_tensor(x, mesh, [Replicate()])
    tr._offset = 0
    out_full = torch.nn.functional.dropout(d_full, p=0.3, training=True)
    full = out_full._local_tensor

    # "sharded" dropout emulated at ws=1 by a manual spec with offsets:
    # fill rows [16:48) as if they were rank 1 of a 4-way Shard(0)
    from vescale_amd.dtensor._dtensor_spec import DTensorSpec
    from vescale_amd.dtensor.placement_types import TensorMeta
    import vescale_amd.ops as ops

    C = ops.require_ext()
    shard = x[16:48].contiguous()
    out, mask = C.philox_dropout(
        shard, [64, 128], [32, 128], [16, 0], 0, False, 777, 0, 0.3, True
    )
    assert torch.equal(out, full[16:48])

    from vescale_amd.dtensor.dispatch import get_dispatcher

    get_dispatcher()._rng_tracker = None


@requires_gpu
def test_gemm8_rot3np_modes():
    """All shipping gemm8 schedules agree with the fp32 reference."""
    import vescale_amd.ops as ops

    C = ops.require_ext()
    torch.manual_seed(3)
    a = torch.randn(512, 256, device="cuda", dtype=torch.bfloat16) / 8
    b = torch.randn(768, 256, device="cuda", dtype=torch.bfloat16) / 8
    ref = a.float() @ b.float().t()
    for mode in (0, 2, 3, 4, 6):
        c = C.gemm_tn8(a, b, mode).float()
        rel = ((c - ref).abs() / ref.abs().clamp_min(1e-2)).max()
        assert float(rel) < 0.05, (mode, float(rel))


@requires_gpu
def test_linear_gemm8_autograd_parity():
    """VESCALE_GEMM=gemm8 linear: fwd + dgrad on the in-tree kernel match
    the library path (wgrad identical by construction — same library op)."""
    from vescale_amd.ops.functional import _Linear8

    torch.manual_seed(5)
    x = (torch.randn(512, 256, device="cuda", dtype=torch.bfloat16) / 8
         ).requires_grad_()
    w = (torch.randn(768, 256, device="cuda", dtype=torch.bfloat16) / 8
         ).requires_grad_()
    y = _Linear8.apply(x, w)
    gy = torch.randn_like(y) / 8
    y.backward(gy)
    xr = x.detach().clone().requires_grad_()
    wr = w.detach().clone().requires_grad_()
    yr = torch.nn.functional.linear(xr, wr)
    yr.backward(gy)
    assert torch.allclose(y.float(), yr.float(), atol=0.05, rtol=0.05)
    assert torch.allclose(x.grad.float(), xr.grad.float(), atol=0.05, rtol=0.05)
    assert torch.allclose(w.grad.float(), wr.grad.float(), atol=0.05, rtol=0.05)
