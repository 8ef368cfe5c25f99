"""Llama-3 model family — MI355X-native implementation.

The flagship bench model (BASELINE.json: Llama-3-8B FSDP tokens/sec).
Forward runs on plain tensors (the FSDP engine manages sharding at the
parameter level); the hot non-GEMM ops are our CDNA4 HIP kernels
(RMSNorm / RoPE / SwiGLU / fused CE), GEMMs go to hipBLASLt via torch.
Parity role: reference examples' HF Llama2 / open_llama configs
(legacy/examples/llama2_4D_finetune/, open_llama_4D_benchmark/).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import (
    build_rope_table,
    linear as vlinear,
    flash_attention_causal,
    fused_add_rmsnorm,
    fused_cross_entropy,
    rmsnorm,
    rope_qkv,
    swiglu_packed,
)
from .llama_tp import TPContext, copy_to_tp, reduce_from_tp


class VLinear(nn.Linear):
    """nn.Linear routed through ops.linear: VESCALE_GEMM=gemm8 runs the
    in-tree 8-phase CDNA4 MFMA kernel (fwd + dgrad) instead of hipBLASLt
    (see ops/functional.py header for the measured A/B and default)."""

    def forward(self, x):
        return vlinear(x, self.weight, self.bias)


@dataclass
class LlamaConfig:
    dim: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    ffn_dim: int = 14336
    vocab_size: int = 128256
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    tie_embeddings: bool = False

    @property
    def head_dim(self) -> int:
        return self.dim // self.n_heads

    def num_params(self) -> int:
        d, f, v = self.dim, self.ffn_dim, self.vocab_size
        per_layer = (
            d * (d + 2 * self.n_kv_heads * self.head_dim)  # wqkv (fused)
            + d * d  # wo
            + 3 * d * f  # w13 (fused gate+up) + w2
            + 2 * d  # norms
        )
        emb = v * d * (1 if self.tie_embeddings else 2)
        return self.n_layers * per_layer + emb + d


def llama3_8b() -> LlamaConfig:
    return LlamaConfig()


def llama3_70b() -> LlamaConfig:
    return LlamaConfig(
        dim=8192, n_layers=80, n_heads=64, n_kv_heads=8, ffn_dim=28672,
        vocab_size=128256,
    )


def llama_tiny(vocab: int = 512, seq: int = 128) -> LlamaConfig:
    return LlamaConfig(
        dim=64, n_layers=2, n_heads=4, n_kv_heads=2, ffn_dim=128,
        vocab_size=vocab, max_seq_len=seq, rope_theta=10000.0,
    )


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: TPContext = None):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or TPContext.single()
        assert cfg.n_heads % self.tp.world == 0 and cfg.n_kv_heads % self.tp.world == 0
        self.hq = cfg.n_heads // self.tp.world
        self.hkv = cfg.n_kv_heads // self.tp.world
        d, hd = cfg.dim, cfg.head_dim
        # fused qkv projection: one GEMM instead of three (TP: local heads)
        self.wqkv = VLinear(d, (self.hq + 2 * self.hkv) * hd, bias=False)
        self.wo = VLinear(self.hq * hd, d, bias=False)
        if self.tp.world > 1:
            # TP metadata for topology-independent (cross-TP reshardable)
            # checkpointing: (dim, global_size, [(local_start, n, global_start)])
            w, r = self.tp.world, self.tp.rank
            q, kv = self.hq * hd, self.hkv * hd
            self.wqkv.weight._tp_shard = (
                0, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd,
                [(0, q, r * q),
                 (q, kv, cfg.n_heads * hd + r * kv),
                 (q + kv, kv, (cfg.n_heads + cfg.n_kv_heads) * hd + r * kv)],
            )
            self.wo.weight._tp_shard = (1, cfg.n_heads * hd, [(0, q, r * q)])

    def forward(self, x: torch.Tensor, rope_table: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        cfg = self.cfg
        x = copy_to_tp(x, self.tp)
        qkv = self.wqkv(x)
        # rope_qkv emits [B, H, S, D] directly (attention's native layout)
        q, k, v = rope_qkv(qkv, rope_table, self.hq, self.hkv, cfg.head_dim)
        # our MFMA flash forward + library backward (ops/functional.py);
        # falls back to SDPA off-GPU or for unsupported shapes
        o = flash_attention_causal(q, k, v)
        o = o.transpose(1, 2).reshape(B, S, self.hq * cfg.head_dim)
        return reduce_from_tp(self.wo(o), self.tp)


class FeedForward(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: TPContext = None):
        super().__init__()
        self.tp = tp or TPContext.single()
        assert cfg.ffn_dim % self.tp.world == 0
        self.ffn_local = cfg.ffn_dim // self.tp.world
        # w1 (gate) and w3 (up) fused into one GEMM (TP: local ffn slice)
        self.w13 = VLinear(cfg.dim, 2 * self.ffn_local, bias=False)
        self.w2 = VLinear(self.ffn_local, cfg.dim, bias=False)
        if self.tp.world > 1:
            fl, r = self.ffn_local, self.tp.rank
            self.w13.weight._tp_shard = (
                0, 2 * cfg.ffn_dim,
                [(0, fl, r * fl), (fl, fl, cfg.ffn_dim + r * fl)],
            )
            self.w2.weight._tp_shard = (1, cfg.ffn_dim, [(0, fl, r * fl)])

    def forward(self, x):
        x = copy_to_tp(x, self.tp)
        return reduce_from_tp(self.w2(swiglu_packed(self.w13(x))), self.tp)


class TransformerBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp: TPContext = None):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = Attention(cfg, tp)
        self.ffn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.ffn = FeedForward(cfg, tp)

    def forward(self, h, delta, rope_table):
        # residual stream (h) with a PENDING delta: the add is fused into
        # the next norm's read pass (ops.fused_add_rmsnorm) — saves a full
        # read+write of the hidden state per residual add
        y, h = fused_add_rmsnorm(
            h if delta is None else delta,
            None if delta is None else h,
            self.attn_norm.weight, self.attn_norm.eps,
        )
        a = self.attn(y, rope_table)
        y2, h = fused_add_rmsnorm(a, h, self.ffn_norm.weight, self.ffn_norm.eps)
        return h, self.ffn(y2)


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_group=None):
        super().__init__()
        self.cfg = cfg
        self.tp = TPContext.from_group(tp_group)
        self.tok_embeddings = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.layers = nn.ModuleList(
            TransformerBlock(cfg, self.tp) for _ in range(cfg.n_layers)
        )
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.output = VLinear(cfg.dim, cfg.vocab_size, bias=False)
        if cfg.tie_embeddings:
            self.output.weight = self.tok_embeddings.weight
        self.register_buffer(
            "rope_table",
            build_rope_table(cfg.max_seq_len, cfg.head_dim, cfg.rope_theta),
            persistent=False,
        )
        self.activation_checkpointing = False

    def init_weights(self, std: float = 0.02):
        for name, p in self.named_parameters():
            if p.ndim >= 2:
                nn.init.normal_(p, mean=0.0, std=std)
            elif "weight" in name:  # norms
                nn.init.ones_(p)
            else:
                nn.init.zeros_(p)

    def forward(
        self,
        tokens: torch.Tensor,
        targets: Optional[torch.Tensor] = None,
        ignore_index: int = -100,
    ):
        h = self.tok_embeddings(tokens)
        tab = self.rope_table
        delta = None
        for layer in self.layers:
            if self.activation_checkpointing and self.training:
                h, delta = torch.utils.checkpoint.checkpoint(
                    layer, h, delta, tab, use_reentrant=False
                )
            else:
                h, delta = layer(h, delta, tab)
        h, _ = fused_add_rmsnorm(
            h if delta is None else delta,
            None if delta is None else h,
            self.norm.weight, self.norm.eps,
        )
        logits = self.output(h)
        if targets is None:
            return logits
        loss = fused_cross_entropy(
            logits.reshape(-1, self.cfg.vocab_size), targets.reshape(-1), ignore_index
        )
        return loss
