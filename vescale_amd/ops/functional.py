"""Autograd wrappers for the CDNA4 HIP kernels, with pure-torch reference
implementations used (a) on CPU boxes for tests, (b) as the fp32 numerics
oracle the GPU tests compare against (SURVEY.md §4 test strategy)."""
from __future__ import annotations

from typing import Optional

import os

import torch


def _ext():
    from . import _C, require_ext

    return require_ext()


def _use_hip(*tensors) -> bool:
    from . import has_ext

    on_gpu = all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if not has_ext():
        # loud failure: GPU present but no extension
        from . import require_ext

        require_ext()
    return True


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------
class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, eps: float):
        if _use_hip(x, weight):
            x2 = x.contiguous()
            out, rrms = _ext().rmsnorm_fwd(x2, weight.contiguous(), eps)
            ctx.save_for_backward(x2, weight, rrms)
            ctx.eps = eps
            ctx.hip = True
            return out
        # reference (fp32 math)
        xf = x.float()
        rrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        out = (xf * rrms * weight.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, rrms.squeeze(-1))
        ctx.eps = eps
        ctx.hip = False
        return out

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w, rrms = ctx.saved_tensors
        if ctx.hip:
            dx, dw = _ext().rmsnorm_bwd(dy.contiguous(), x, w, rrms)
            return dx, dw.to(w.dtype), None
        xf = x.float()
        dyf = dy.float()
        wf = w.float()
        r = rrms.unsqueeze(-1)
        H = x.shape[-1]
        dot = (dyf * wf * xf).sum(-1, keepdim=True)
        dx = (r * wf * dyf - xf * dot * r.pow(3) / H).to(x.dtype)
        dw = (dyf * xf * r).reshape(-1, H).sum(0).to(w.dtype)
        return dx, dw, None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    return _RMSNorm.apply(x, weight, eps)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------
def build_rope_table(
    seq_len: int, head_dim: int, theta: float = 500000.0, device="cpu"
) -> torch.Tensor:
    """[S, D/2, 2] fp32 (cos, sin) — host-precomputed (guide App. B: no
    on-device trig)."""
    rot = head_dim // 2
    # explicit cpu device so meta-device init contexts don't capture these
    inv = 1.0 / (theta ** (torch.arange(0, rot, dtype=torch.float64, device="cpu") / rot))
    pos = torch.arange(seq_len, dtype=torch.float64, device="cpu")
    ang = torch.outer(pos, inv)
    tab = torch.stack([ang.cos(), ang.sin()], dim=-1).to(torch.float32)
    return tab.to(device)


class _RoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, table: torch.Tensor, pos_offset: int):
        ctx.pos_offset = pos_offset
        if _use_hip(x, table):
            ctx.hip = True
            ctx.save_for_backward(table)
            return _ext().rope(x.contiguous(), table, pos_offset, False, False)
        ctx.hip = False
        ctx.save_for_backward(table)
        return _rope_ref(x, table, pos_offset, False)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (table,) = ctx.saved_tensors
        if ctx.hip:
            return _ext().rope(dy.contiguous(), table, ctx.pos_offset, True, False), None, None
        return _rope_ref(dy, table, ctx.pos_offset, True), None, None


def _rope_ref(x, table, pos_offset, backward):
    B, S, H, D = x.shape
    rot = D // 2
    tb = table[pos_offset : pos_offset + S]  # [S, rot, 2]
    cos = tb[..., 0].unsqueeze(0).unsqueeze(2)  # [1,S,1,rot]
    sin = tb[..., 1].unsqueeze(0).unsqueeze(2)
    if backward:
        sin = -sin
    xf = x.float()
    x0, x1 = xf[..., :rot], xf[..., rot:]
    o0 = x0 * cos - x1 * sin
    o1 = x1 * cos + x0 * sin
    return torch.cat([o0, o1], dim=-1).to(x.dtype)


class _FusedAddRMSNorm(torch.autograd.Function):
    """Residual-stream add fused into RMSNorm: (x, res) -> (norm(x+res),
    x+res).  res may be None (start of the stream).  Backward fuses the
    residual grad accumulation into the norm backward (kernel dres path).
    The grad w.r.t. x and res is the same tensor (add distributes)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, res, weight: torch.Tensor, eps: float):
        ctx.has_res = res is not None
        if _use_hip(x, weight) and (res is None or res.dtype == x.dtype):
            out, res_new, rrms = _ext().rmsnorm_res_fwd(
                x.contiguous(), res.contiguous() if res is not None else None,
                weight.contiguous(), eps,
            )
            ctx.save_for_backward(res_new, weight, rrms)
            ctx.hip = True
            return out, res_new
        res_new = x if res is None else (x + res)
        xf = res_new.float()
        rrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        out = (xf * rrms * weight.float()).to(x.dtype)
        ctx.save_for_backward(res_new, weight, rrms.squeeze(-1))
        ctx.hip = False
        return out, res_new

    @staticmethod
    def backward(ctx, dy: torch.Tensor, dres_new):
        res_new, w, rrms = ctx.saved_tensors
        if ctx.hip:
            dx, dw = _ext().rmsnorm_bwd(
                dy.contiguous(), res_new, w, rrms,
                dres_new.contiguous() if dres_new is not None else None,
            )
            dw = dw.to(w.dtype)
        else:
            xf = res_new.float()
            dyf = dy.float()
            r = rrms.unsqueeze(-1)
            wf = w.float()
            h = res_new.shape[-1]
            dot = (dyf * wf * xf).sum(-1, keepdim=True)
            dx = (r * wf * dyf - xf * (dot * r.pow(3) / h)).to(res_new.dtype)
            dw = (dyf * xf * r).reshape(-1, h).sum(0).to(w.dtype)
            if dres_new is not None:
                dx = dx + dres_new
        dres = dx if ctx.has_res else None
        return dx, dres, dw, None


def fused_add_rmsnorm(x, res, weight, eps):
    """(norm(x + res), x + res); res=None starts the residual stream."""
    return _FusedAddRMSNorm.apply(x, res, weight, eps)


def rope_apply(x: torch.Tensor, table: torch.Tensor, pos_offset: int = 0) -> torch.Tensor:
    """x: [B, S, H, D]."""
    return _RoPE.apply(x, table, pos_offset)


class _RoPEQKV(torch.autograd.Function):
    """Fused qkv-split + RoPE (packed wqkv GEMM output -> q/k/v)."""

    @staticmethod
    def forward(ctx, qkv: torch.Tensor, table: torch.Tensor, n_heads: int,
                n_kv_heads: int, head_dim: int, pos_offset: int):
        B, S, _ = qkv.shape
        ctx.dims = (B, S, n_heads, n_kv_heads, head_dim, pos_offset)
        ctx.save_for_backward(table)
        if _use_hip(qkv, table):
            ctx.hip = True
            q, k, v = _ext().rope_qkv_fwd(
                qkv.contiguous(), table, B, S, n_heads, n_kv_heads, head_dim,
                pos_offset,
            )
            return q, k, v
        ctx.hip = False
        d = n_heads * head_dim
        kv = n_kv_heads * head_dim
        q = qkv[..., :d].reshape(B, S, n_heads, head_dim)
        k = qkv[..., d : d + kv].reshape(B, S, n_kv_heads, head_dim)
        v = qkv[..., d + kv :].reshape(B, S, n_kv_heads, head_dim)
        q = _rope_ref(q, table, pos_offset, False)
        k = _rope_ref(k, table, pos_offset, False)
        # match the HIP kernel's [B, H, S, D] output layout
        return (
            q.permute(0, 2, 1, 3).contiguous(),
            k.permute(0, 2, 1, 3).contiguous(),
            v.permute(0, 2, 1, 3).contiguous(),
        )

    @staticmethod
    def backward(ctx, dq, dk, dv):
        (table,) = ctx.saved_tensors
        B, S, Hq, Hkv, D, pos = ctx.dims
        if ctx.hip:
            dqkv = _ext().rope_qkv_bwd(
                dq.contiguous(), dk.contiguous(), dv.contiguous(),
                table, B, S, Hq, Hkv, D, pos,
            )
            return dqkv, None, None, None, None, None
        dq = dq.permute(0, 2, 1, 3)  # [B,H,S,D] -> [B,S,H,D]
        dk = dk.permute(0, 2, 1, 3)
        dv = dv.permute(0, 2, 1, 3)
        dqr = _rope_ref(dq, table, pos, True).reshape(B, S, Hq * D)
        dkr = _rope_ref(dk, table, pos, True).reshape(B, S, Hkv * D)
        dvr = dv.reshape(B, S, Hkv * D)
        return torch.cat([dqr, dkr, dvr], dim=-1), None, None, None, None, None


def rope_qkv(qkv: torch.Tensor, table: torch.Tensor, n_heads: int,
             n_kv_heads: int, head_dim: int, pos_offset: int = 0):
    """qkv: [B, S, (Hq+2*Hkv)*D] packed -> (q, k, v) in [B, H, S, D] (the
    attention kernels' native layout), RoPE applied to q and k."""
    return _RoPEQKV.apply(qkv, table, n_heads, n_kv_heads, head_dim, pos_offset)


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------
class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate: torch.Tensor, up: torch.Tensor):
        if _use_hip(gate, up):
            g = gate.contiguous()
            u = up.contiguous()
            ctx.save_for_backward(g, u)
            ctx.hip = True
            return _ext().swiglu_fwd(g, u)
        ctx.save_for_backward(gate, up)
        ctx.hip = False
        return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        gate, up = ctx.saved_tensors
        if ctx.hip:
            dgate, dup = _ext().swiglu_bwd(dy.contiguous(), gate, up)
            return dgate, dup
        g = gate.float()
        u = up.float()
        d = dy.float()
        sig = torch.sigmoid(g)
        dgate = (d * u * sig * (1 + g * (1 - sig))).to(gate.dtype)
        dup = (d * g * sig).to(up.dtype)
        return dgate, dup


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return _SwiGLU.apply(gate, up)


class _SwiGLUPacked(torch.autograd.Function):
    """gu = [.., 2F] (gate||up) -> [.., F]; avoids split+cat round trips."""

    @staticmethod
    def forward(ctx, gu: torch.Tensor):
        if _use_hip(gu):
            g = gu.contiguous()
            ctx.save_for_backward(g)
            ctx.hip = True
            return _ext().swiglu_packed_fwd(g)
        ctx.save_for_backward(gu)
        ctx.hip = False
        F = gu.shape[-1] // 2
        gate, up = gu[..., :F], gu[..., F:]
        return (torch.nn.functional.silu(gate.float()) * up.float()).to(gu.dtype)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (gu,) = ctx.saved_tensors
        if ctx.hip:
            return _ext().swiglu_packed_bwd(dy, gu)
        F = gu.shape[-1] // 2
        g = gu[..., :F].float()
        u = gu[..., F:].float()
        d = dy.float()
        sig = torch.sigmoid(g)
        dgate = d * u * sig * (1 + g * (1 - sig))
        dup = d * g * sig
        return torch.cat([dgate, dup], dim=-1).to(gu.dtype)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    return _SwiGLUPacked.apply(gu)


# ---------------------------------------------------------------------------
# Fused cross-entropy (mean over non-ignored tokens)
# ---------------------------------------------------------------------------
class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor, ignore_index: int):
        n_tok = (target != ignore_index).sum().clamp(min=1)
        if _use_hip(logits):
            lg = logits.contiguous()
            loss, lse = _ext().ce_fwd(lg, target.contiguous(), ignore_index)
            ctx.save_for_backward(lg, target, lse, n_tok)
            ctx.ignore_index = ignore_index
            ctx.hip = True
            return loss.sum() / n_tok.to(loss.dtype)
        xf = logits.float()
        lse = torch.logsumexp(xf, dim=-1)
        valid = target != ignore_index
        tgt = target.clamp(min=0)
        xt = xf.gather(-1, tgt.unsqueeze(-1)).squeeze(-1)
        loss = torch.where(valid, lse - xt, torch.zeros_like(lse))
        ctx.save_for_backward(logits, target, lse, n_tok)
        ctx.ignore_index = ignore_index
        ctx.hip = False
        return loss.sum() / n_tok.to(loss.dtype)

    @staticmethod
    def backward(ctx, dloss: torch.Tensor):
        logits, target, lse, n_tok = ctx.saved_tensors
        scale = (dloss / n_tok.to(dloss.dtype)).float()
        if ctx.hip:
            drow = scale.expand(lse.shape[0]).contiguous()
            dlogits = _ext().ce_bwd(
                logits, target, lse, drow, ctx.ignore_index, False
            )
            return dlogits, None, None
        xf = logits.float()
        p = torch.exp(xf - lse.unsqueeze(-1))
        valid = (target != ctx.ignore_index).unsqueeze(-1)
        tgt = target.clamp(min=0)
        p.scatter_add_(
            -1, tgt.unsqueeze(-1), -torch.ones_like(tgt, dtype=p.dtype).unsqueeze(-1)
        )
        d = (p * scale * valid).to(logits.dtype)
        return d, None, None


def fused_cross_entropy(
    logits: torch.Tensor, target: torch.Tensor, ignore_index: int = -100
) -> torch.Tensor:
    """logits [N, V] (any leading dims flattened by caller), target [N]."""
    return _FusedCE.apply(logits, target, ignore_index)


# ---------------------------------------------------------------------------
# Flash attention: library forward (returns logsumexp) + our CDNA4 MFMA
# backward (ops/csrc/attention_bwd.hip) — the stock backward is the profiled
# bottleneck (~255 TF effective); causal, head_dim 128, GQA.
# ---------------------------------------------------------------------------
class _FlashAttention(torch.autograd.Function):
    """Causal flash attention, fully on our MFMA kernels: fa_fwd
    (attention_fwd.hip, ~565 TF at the Llama-8B shape vs the stock AOTriton
    forward's ~477 TF) + fa_bwd2 (attention_bwd2.hip, ~18.7 ms vs the
    library backward's ~20.9 ms at that shape).  Set VESCALE_FA=aten to
    fall back to the torch/AOTriton kernels."""

    @staticmethod
    def forward(ctx, q, k, v, scale):
        # q,k,v: [B, H, S, D] contiguous bf16
        out, lse = _ext().fa_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        # default OFF: the fused dv+dk kernel is numerically exact but
        # measured SLOWER (whole bwd 17.9 ms vs 11.2 ms at B4 H32/8 S8192,
        # gpurun_out/fa_fused_check.log) — its 2-accumulator register set
        # only fits at 1 wave/SIMD (4-wave blocks), and the lost latency
        # hiding outweighs the deduplicated ST GEMM + staging.  Kept for A/B.
        fused = os.environ.get("VESCALE_FA_FUSED_DVDK", "0") == "1"
        dq, dk, dv = _ext().fa_bwd2(q, k, v, out, dout, lse, ctx.scale, fused)
        return dq, dk, dv, None


def flash_attention_causal(q, k, v):
    """Causal flash attention, [B,H,S,D] bf16; GQA via Hkv < Hq.  Falls back
    to torch SDPA off-GPU / without the extension loudly on GPU."""
    import math

    scale = 1.0 / math.sqrt(q.shape[-1])
    if (
        _use_hip(q, k, v)
        and q.shape[-1] == 128
        and q.shape[2] % 256 == 0
        and os.environ.get("VESCALE_FA", "hip") == "hip"
    ):
        return _FlashAttention.apply(
            q.contiguous(), k.contiguous(), v.contiguous(), scale
        )
    return torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True, enable_gqa=q.shape[1] != k.shape[1]
    )


# ---------------------------------------------------------------------------
# flat AdamW + grad utilities (no autograd)
# ---------------------------------------------------------------------------
@torch.no_grad()
def adamw_step_flat(
    param_bf16: torch.Tensor,
    master_f32: Optional[torch.Tensor],
    grad: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    *,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.95,
    eps: float = 1e-8,
    weight_decay: float = 0.0,
    step: int = 1,
    grad_scale: float = 1.0,
    clip_scale: Optional[torch.Tensor] = None,
):
    if _use_hip(param_bf16, grad, m, v):
        _ext().adamw_step(
            param_bf16, master_f32, grad, m, v, lr, beta1, beta2, eps,
            weight_decay, step, grad_scale, clip_scale,
        )
        return
    g = grad.float() * grad_scale
    if clip_scale is not None:
        g = g * clip_scale.item()
    p = master_f32 if master_f32 is not None else param_bf16.float()
    p.mul_(1 - lr * weight_decay)
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (v / bc2).sqrt().add_(eps)
    p.addcdiv_(m, denom, value=-lr / bc1)
    param_bf16.copy_(p.to(param_bf16.dtype))


@torch.no_grad()
def l2norm_sq(x: torch.Tensor) -> torch.Tensor:
    if _use_hip(x):
        return _ext().l2norm_sq(x).squeeze(0)
    return x.float().pow(2).sum()


@torch.no_grad()
def scale_flat_(x: torch.Tensor, scale) -> None:
    if _use_hip(x):
        if isinstance(scale, torch.Tensor):
            _ext().scale_(x, scale.float().contiguous(), 1.0)
        else:
            _ext().scale_(x, None, float(scale))
        return
    if isinstance(scale, torch.Tensor):
        x.mul_(scale.to(x.dtype))
    else:
        x.mul_(scale)


# ---------------------------------------------------------------------------
# Linear (in-tree CDNA4 GEMM path)
# ---------------------------------------------------------------------------
# VESCALE_GEMM selects the nn.Linear GEMM backend:
#   "blaslt" (default) — torch.matmul -> hipBLASLt Tensile assembly kernels.
#   "gemm8"            — the in-tree 256x256 8-phase MFMA kernel
#                        (ops/csrc/gemm8.hip, rot3np schedule) for the
#                        forward and dgrad GEMMs; wgrad (both operands
#                        contraction-major) stays on the library until the
#                        tr-read wgrad kernel lands.  Honest A/B:
#                        profiles/gemm8_variants_ab_r2.log — the in-tree
#                        kernel is ~75% of hipBLASLt on Llama shapes, so
#                        blaslt remains the default (VERDICT r1 item 1:
#                        "do not ship a regression").
# Reference parity: sharded matmul family, SURVEY.md §2.7;
# legacy/vescale/dtensor/ops/matrix_ops.py:113.
def _gemm_backend() -> str:
    return os.environ.get("VESCALE_GEMM", "blaslt")


def _g8_ok(m: int, n: int, k: int) -> bool:
    return m % 256 == 0 and n % 256 == 0 and k % 64 == 0


class _Linear8(torch.autograd.Function):
    """y = x @ w^T via the in-tree TN kernel; dgrad = gemm_tn(dy, w^T)
    (one cheap [N,K]->[K,N] transpose per backward — arithmetic intensity
    of the GEMM is ~4 orders above the copy); wgrad on the library."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor):
        ctx.save_for_backward(x, w)
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        m, k = x2.shape
        n = w.shape[0]
        if x.is_cuda and x.dtype == torch.bfloat16 and _g8_ok(m, n, k):
            y = _ext().gemm_tn8(x2.contiguous(), w.contiguous(), 4)
        else:
            y = x2 @ w.t()
        return y.reshape(*xs[:-1], n)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w = ctx.saved_tensors
        dys = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        m, n = dys.shape
        k = w.shape[1]
        dx = dw = None
        if ctx.needs_input_grad[0]:
            if dy.is_cuda and dy.dtype == torch.bfloat16 and _g8_ok(m, k, n):
                wt = w.t().contiguous()
                dx = _ext().gemm_tn8(dys.contiguous(), wt, 4)
            else:
                dx = dys @ w
            dx = dx.reshape(x.shape)
        if ctx.needs_input_grad[1]:
            dw = dys.t() @ x2  # library wgrad (see header note)
        return dx, dw


def linear(x: torch.Tensor, w: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    if _gemm_backend() == "gemm8" and bias is None:
        return _Linear8.apply(x, w)
    return torch.nn.functional.linear(x, w, bias)
