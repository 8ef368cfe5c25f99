"""Mixtral (MoE) model family — MI355X-native.

Parity role: reference examples legacy/examples/mixtral_4D_benchmark/ and
mixtral_EP_training/ — top-k routed sparse FFN; EP runs through
vescale_amd.moe.parallelize_experts (token all-to-all over xGMI).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..moe._utils import global_all_to_all_single
from ..ops import build_rope_table, fused_cross_entropy, rmsnorm, rope_qkv, swiglu_packed
from .llama import Attention, LlamaConfig, RMSNorm


@dataclass
class MixtralConfig(LlamaConfig):
    n_experts: int = 8
    top_k: int = 2
    router_aux_loss_coef: float = 0.0


def mixtral_8x7b() -> MixtralConfig:
    return MixtralConfig(
        dim=4096, n_layers=32, n_heads=32, n_kv_heads=8, ffn_dim=14336,
        vocab_size=32000, rope_theta=1e6, n_experts=8, top_k=2,
    )


def mixtral_tiny(vocab: int = 256, seq: int = 64) -> MixtralConfig:
    return MixtralConfig(
        dim=32, n_layers=2, n_heads=4, n_kv_heads=2, ffn_dim=64,
        vocab_size=vocab, max_seq_len=seq, rope_theta=10000.0,
        n_experts=4, top_k=2,
    )


class Expert(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.w13 = nn.Linear(cfg.dim, 2 * cfg.ffn_dim, bias=False)
        self.w2 = nn.Linear(cfg.ffn_dim, cfg.dim, bias=False)

    def forward(self, x):
        return self.w2(swiglu_packed(self.w13(x)))


class MoELayer(nn.Module):
    """Top-k routed MoE.  Dense on one rank; expert-parallel after
    vescale_amd.moe.parallelize_experts()."""

    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.router = nn.Linear(cfg.dim, cfg.n_experts, bias=False)
        self.experts = nn.ModuleList(Expert(cfg) for _ in range(cfg.n_experts))
        self._ep = False
        self.ep_group = None
        self.allocator = None
        self.dispatcher = None
        self.local_expert_ids: List[int] = []

    def make_expert(self) -> nn.Module:
        """Fresh expert on this layer's device (dynamic re-placement)."""
        dev = self.router.weight.device
        dt = self.router.weight.dtype
        return Expert(self.cfg).to(device=dev, dtype=dt)

    def enable_expert_parallel(self, group, allocator, dispatcher, local_ids):
        self._ep = True
        self.ep_group = group
        self.allocator = allocator
        self.dispatcher = dispatcher
        self.local_expert_ids = list(local_ids)

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, D = x.shape
        xt = x.reshape(-1, D)
        logits = self.router(xt)  # [T, E]
        weights, ids = torch.topk(logits, self.cfg.top_k, dim=-1)
        weights = torch.softmax(weights.float(), dim=-1).to(x.dtype)  # [T, k]
        if not self._ep:
            out = self._dense_moe(xt, weights, ids)
        else:
            out = self._ep_moe(xt, weights, ids)
        return out.reshape(B, S, D)

    def _dense_moe(self, xt, weights, ids):
        T, D = xt.shape
        k = self.cfg.top_k
        flat_ids = ids.reshape(-1)          # [T*k]
        flat_x = xt.repeat_interleave(k, dim=0)
        order = torch.argsort(flat_ids, stable=True)
        counts = torch.bincount(flat_ids, minlength=self.cfg.n_experts)
        sorted_x = flat_x[order]
        outs = torch.empty_like(sorted_x)
        off = 0
        for e in range(self.cfg.n_experts):
            n = int(counts[e])
            if n:
                outs[off : off + n] = self.experts[e](sorted_x[off : off + n])
            off += n
        inv = torch.empty_like(order)
        inv[order] = torch.arange(order.numel(), device=order.device)
        unsorted = outs[inv].reshape(T, k, D)
        return (unsorted * weights.unsqueeze(-1)).sum(dim=1)

    def _ep_moe(self, xt, weights, ids):
        """Expert-parallel path: sort by (owner, expert) -> uneven
        all_to_all -> local expert compute -> reverse."""
        T, D = xt.shape
        k = self.cfg.top_k
        E = self.cfg.n_experts
        W = self.allocator.ep_world
        flat_ids = ids.reshape(-1)
        flat_x = xt.repeat_interleave(k, dim=0)
        perm, send_splits, sorted_experts = self.dispatcher.dispatch_plan(flat_ids)
        xs = flat_x[perm]
        # exchange split sizes + per-expert counts in one int tensor
        counts = torch.zeros(W * E, dtype=torch.int64)
        owner = torch.tensor([self.allocator.owner_of(e) for e in range(E)])
        cnt = torch.bincount(flat_ids.cpu(), minlength=E)
        for e in range(E):
            counts[owner[e] * E + e] = cnt[e]
        recv_counts = torch.empty_like(counts)
        _a2a_int(recv_counts, counts, self.ep_group, W, E)
        recv_splits = [int(recv_counts[r * E : (r + 1) * E].sum()) for r in range(W)]
        recv = global_all_to_all_single(xs, recv_splits, send_splits, self.ep_group)
        # group received tokens by expert: each src chunk is expert-sorted
        recv_expert_ids = []
        for r in range(W):
            for e in self.local_expert_ids:
                n = int(recv_counts[r * E + e])
                if n:
                    recv_expert_ids.append(torch.full((n,), e, dtype=torch.int64))
        if recv_expert_ids:
            rei = torch.cat(recv_expert_ids).to(recv.device)
        else:
            rei = torch.zeros(0, dtype=torch.int64, device=recv.device)
        order2 = torch.argsort(rei, stable=True)
        grouped = recv[order2]
        outs = torch.empty_like(grouped)
        off = 0
        for e in self.local_expert_ids:
            n = int(sum(int(recv_counts[r * E + e]) for r in range(W)))
            if n:
                outs[off : off + n] = self.experts[e](grouped[off : off + n])
            off += n
        inv2 = torch.empty_like(order2)
        inv2[order2] = torch.arange(order2.numel(), device=order2.device)
        back = outs[inv2]
        # reverse all_to_all
        combined = global_all_to_all_single(back, send_splits, recv_splits, self.ep_group)
        inv = torch.empty_like(perm)
        inv[perm] = torch.arange(perm.numel(), device=perm.device)
        unsorted = combined[inv].reshape(T, k, D)
        return (unsorted * weights.unsqueeze(-1)).sum(dim=1)


def _a2a_int(recv, send, group, W, E):
    if group is None or not dist.is_initialized() or dist.get_world_size(group) == 1:
        recv.copy_(send)
        return
    try:
        dist.all_to_all_single(recv, send, group=group)
    except RuntimeError:
        from ..moe._utils import _a2a_fallback

        _a2a_fallback(recv.view(W, E), send.view(W, E), [1] * W, [1] * W, group)


class MixtralBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.ffn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.moe = MoELayer(cfg)

    def forward(self, x, rope_table):
        x = x + self.attn(self.attn_norm(x), rope_table)
        x = x + self.moe(self.ffn_norm(x))
        return x


class MixtralModel(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_embeddings = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.layers = nn.ModuleList(MixtralBlock(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.output = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        self.register_buffer(
            "rope_table",
            build_rope_table(cfg.max_seq_len, cfg.head_dim, cfg.rope_theta),
            persistent=False,
        )

    def init_weights(self, std: float = 0.02):
        for name, p in self.named_parameters():
            if p.ndim >= 2:
                nn.init.normal_(p, mean=0.0, std=std)
            elif "weight" in name:
                nn.init.ones_(p)
            else:
                nn.init.zeros_(p)

    def forward(self, tokens, targets=None, ignore_index: int = -100):
        h = self.tok_embeddings(tokens)
        for layer in self.layers:
            h = layer(h, self.rope_table)
        h = self.norm(h)
        logits = self.output(h)
        if targets is None:
            return logits
        return fused_cross_entropy(
            logits.reshape(-1, self.cfg.vocab_size), targets.reshape(-1), ignore_index
        )
