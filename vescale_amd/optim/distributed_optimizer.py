"""DistributedOptimizer — ZeRO-2+ optimizer-state sharding over the DDP
flat grad buffer.

Parity: legacy/vescale/optim/distributed_optimizer.py:131-1290 — the grad
buffer is sharded EVENLY across DP ranks ignoring parameter boundaries
(range maps); fp32 master slices live only on their owner rank; step =
copy grads (zero-copy views into the fp32 buffer) -> clip -> inner
optimizer on slices -> write to the param buffer -> all-gather param
ranges back.  OptimizerStateSpec maps each 1-D flat range back to global
tensor coordinates for checkpoint resharding.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from ..dtensor import DTensor
from .clip_grads import clip_grad_norm_fp32


@dataclass
class OptimizerStateSpec:
    """Where a rank's flat shard of one parameter sits globally (reference
    distributed_optimizer.py:51)."""

    fqn: str
    global_shape: Tuple[int, ...]
    local_flat_start: int  # offset within the param's LOCAL flat extent
    local_flat_end: int
    # the param's own (DTensor) placements apply above this flat range


class DistributedOptimizer:
    def __init__(
        self,
        optimizer: torch.optim.Optimizer,
        models: Sequence[nn.Module],
        *,
        clip_grad: float = 0.0,
        overlap_param_gather: bool = False,
        extra_norm_pgs: Optional[List] = None,
    ):
        self.optimizer = optimizer
        self.models = list(models)
        self.clip_grad = clip_grad
        self.overlap_param_gather = overlap_param_gather
        self.extra_norm_pgs = extra_norm_pgs or []

        m0 = self.models[0]
        self.dp_group = m0.dp_group
        self.dp_world = m0.dp_world
        self.dp_rank = (
            dist.get_rank(self.dp_group)
            if self.dp_group is not None and dist.is_initialized()
            else 0
        )

        # fqn lookup for state specs
        self._fqn_of: Dict[int, str] = {}
        for m in self.models:
            target = m.module if hasattr(m, "module") else m
            for name, p in target.named_parameters():
                self._fqn_of[id(p)] = name

        # build range maps + master slices per grad buffer.
        # CRITICAL: ranges are PER BUCKET, not one contiguous range over the
        # whole buffer — GradBuffer reduce-scatters each bucket independently,
        # so this rank's reduced grads land at
        # [bucket.offset + rank*shard, bucket.offset + (rank+1)*shard) within
        # EVERY bucket (reference build_model_gbuf_range uses bucket.offset the
        # same way, legacy/vescale/optim/distributed_optimizer.py:399-600).
        # entries: (model, gbuf, param, pstart, pend, istart, iend, main)
        self._slices = []
        self.param_buffers = []  # (model, gbuf, pbuf, [(off, size, r0, r1)])
        for m in self.models:
            for dt, gb in m.grad_buffers.items():
                pdtype = None
                bucket_ranges = []  # (bucket_off, bucket_size, r0, r1)
                for b in gb.buckets:
                    bn = b.data.numel()
                    if self.dp_world <= 1:
                        r0, r1 = b.offset, b.offset + bn
                    else:
                        s = bn // self.dp_world
                        r0 = b.offset + self.dp_rank * s
                        r1 = r0 + s
                    bucket_ranges.append((b.offset, bn, r0, r1))
                    for p in sorted(
                        b.params, key=lambda q: gb.param_index_map[q][0]
                    ):
                        ps, pe = gb.param_index_map[p]
                        pdtype = _local(p).dtype
                        i0, i1 = max(ps, r0), min(pe, r1)
                        if i0 >= i1:
                            continue
                        lf = _local(p).reshape(-1)
                        main = lf[i0 - ps : i1 - ps].detach().float().clone()
                        self._slices.append((m, gb, p, ps, pe, i0, i1, main))
                pbuf = torch.empty(
                    gb.numel, dtype=pdtype or torch.float32, device=gb.data.device
                )
                self.param_buffers.append((m, gb, pbuf, bucket_ranges))

        # rebuild inner-optimizer param groups over the master slices
        mains_of_param: Dict[int, List[torch.Tensor]] = {}
        for (_, _, p, _, _, _, _, main) in self._slices:
            mains_of_param.setdefault(id(p), []).append(main)
        new_groups = []
        for group in optimizer.param_groups:
            g = {k: v for k, v in group.items() if k != "params"}
            g["params"] = [
                main for p in group["params"] for main in mains_of_param.get(id(p), [])
            ]
            new_groups.append(g)
        optimizer.param_groups = []
        for g in new_groups:
            optimizer.add_param_group(g)

    # ------------------------------------------------------------------
    def _copy_model_grads_to_main_grads(self):
        for (m, gb, p, ps, pe, i0, i1, main) in self._slices:
            # the fp32 grad buffer range IS the grad (zero-copy view)
            main.grad = gb.data.narrow(0, i0, i1 - i0)
            if main.grad.dtype != torch.float32:
                main.grad = main.grad.float()

    def _copy_main_params_to_model_params(self):
        # write master -> param buffer, gather per bucket, scatter to params
        for (m, gb, pbuf, bucket_ranges) in self.param_buffers:
            # fill this rank's owned ranges from the masters
            for (m2, gb2, p, ps, pe, i0, i1, main) in self._slices:
                if gb2 is not gb:
                    continue
                pbuf.narrow(0, i0, i1 - i0).copy_(main.to(pbuf.dtype))
            if self.dp_world > 1:
                # gather each bucket: rank r's shard sits at off + r*shard
                for (off, size, r0, r1) in bucket_ranges:
                    bview = pbuf.narrow(0, off, size)
                    shard = pbuf.narrow(0, r0, r1 - r0).clone()
                    try:
                        dist.all_gather_into_tensor(
                            bview, shard, group=self.dp_group
                        )
                    except RuntimeError:
                        bufs = list(bview.chunk(self.dp_world))
                        dist.all_gather(bufs, shard, group=self.dp_group)
            # copy back into param locals
            for p, (ps, pe) in gb.param_index_map.items():
                _local(p).reshape(-1).copy_(pbuf.narrow(0, ps, pe - ps))

    # ------------------------------------------------------------------
    @torch.no_grad()
    def step(self):
        for m in self.models:
            m.finish_grad_sync()
        self._copy_model_grads_to_main_grads()
        norm = None
        if self.clip_grad > 0:
            grads = [s[7].grad for s in self._slices]
            pgs = [self.dp_group] + list(self.extra_norm_pgs)
            norm = clip_grad_norm_fp32(grads, self.clip_grad, pgs)
        self.optimizer.step()
        self._copy_main_params_to_model_params()
        return norm

    def zero_grad(self, set_to_none: bool = True):
        for m in self.models:
            m.zero_grad_buffer()
        for s in self._slices:
            s[7].grad = None

    # ------------------------------------------------------------------
    def state_specs(self) -> List[OptimizerStateSpec]:
        out = []
        for (m, gb, p, ps, pe, i0, i1, main) in self._slices:
            d = p.data
            shape = tuple(d.shape)
            out.append(
                OptimizerStateSpec(
                    fqn=self._fqn_of.get(id(p), "?"),
                    global_shape=shape,
                    local_flat_start=i0 - ps,
                    local_flat_end=i1 - ps,
                )
            )
        return out

    def state_dict(self):
        """Flat-range state dict: per slice, the fp32 master + inner state,
        keyed by (fqn, local range) — reshardable on load."""
        inner = self.optimizer.state_dict()
        slices = []
        for idx, (m, gb, p, ps, pe, i0, i1, main) in enumerate(self._slices):
            st = self.optimizer.state.get(main, {})
            slices.append(
                {
                    "fqn": self._fqn_of.get(id(p), "?"),
                    "start": i0 - ps,
                    "end": i1 - ps,
                    "master": main,
                    "state": {k: v for k, v in st.items()},
                }
            )
        return {"slices": slices, "param_group_hyper": [
            {k: v for k, v in g.items() if k != "params"}
            for g in self.optimizer.param_groups
        ]}

    def load_state_dict(self, sd):
        by_key = {(s["fqn"], s["start"], s["end"]): s for s in sd["slices"]}
        for (m, gb, p, ps, pe, i0, i1, main) in self._slices:
            key = (self._fqn_of.get(id(p), "?"), i0 - ps, i1 - ps)
            if key in by_key:
                s = by_key[key]
                main.copy_(s["master"])
                if s["state"]:
                    self.optimizer.state[main] = {
                        k: (v.clone() if isinstance(v, torch.Tensor) else v)
                        for k, v in s["state"].items()
                    }

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    # -- reference API parity (distributed_optimizer.py:1223-1276) -----
    def get_parameters(self) -> List[torch.Tensor]:
        """The fp32 master shards this rank optimizes."""
        return [s[7] for s in self._slices]

    def get_main_grads_for_grad_norm(self) -> List[torch.Tensor]:
        return [s[7].grad for s in self._slices if s[7].grad is not None]

    def clip_grad_norm(self, clip_grad: float):
        """One-off clip of the current main grads (step() already clips
        when constructed with clip_grad>0)."""
        pgs = [self.dp_group] + list(self.extra_norm_pgs)
        return clip_grad_norm_fp32(
            self.get_main_grads_for_grad_norm(), clip_grad, pgs
        )


def _local(p) -> torch.Tensor:
    d = p.data
    if isinstance(d, DTensor) or hasattr(d, "_local_tensor"):
        return d._local_tensor
    return d
