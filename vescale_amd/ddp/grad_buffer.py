"""Flat gradient buffer with bucketed, overlapped reduction.

Parity: legacy/vescale/ddp/grad_buffer.py (GradBuffer :226, Bucket :27,
start_grad_sync :114) — Megatron-lineage flat-buffer DDP redesigned for
RCCL/xGMI: bucket size defaults to 40 MB-class so each collective is
bandwidth-bound on the per-link 153 GB/s xGMI path rather than
latency-bound.
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist


def _alloc_aligned(n: int, align: int) -> int:
    return (n + align - 1) // align * align


class Bucket:
    def __init__(
        self,
        params: List[torch.nn.Parameter],
        data: torch.Tensor,
        offset: int,
        dp_group,
        dp_world: int,
        use_distributed_optimizer: bool,
        overlap_grad_reduce: bool,
    ):
        self.params = set(params)
        self.params_with_grad = set()
        self.data = data  # view of the flat buffer
        self.offset = offset
        self.dp_group = dp_group
        self.dp_world = dp_world
        self.use_distributed_optimizer = use_distributed_optimizer
        self.overlap_grad_reduce = overlap_grad_reduce
        self.comm_handle = None
        self.comm_issued = False
        # Megatron-style flag: with gradient accumulation (PP microbatches,
        # zero-bubble W-phases) hooks fire once per microbatch; sync must only
        # start on the LAST one.  Default True = one backward per step.
        self.is_last_microbatch = True

    def reset(self):
        self.params_with_grad = set()
        self.comm_handle = None
        self.comm_issued = False

    def shard_size(self) -> int:
        return self.data.numel() // self.dp_world

    def start_grad_sync(self):
        assert not self.comm_issued
        self.comm_issued = True
        if self.dp_world == 1 or self.dp_group is None:
            return
        self.data.div_(self.dp_world)
        if self.use_distributed_optimizer:
            # reduce-scatter into this rank's range (ZeRO grad sharding)
            rank = dist.get_rank(self.dp_group)
            s = self.shard_size()
            out = self.data.narrow(0, rank * s, s)
            try:
                self.comm_handle = dist.reduce_scatter_tensor(
                    out, self.data, group=self.dp_group, async_op=self.overlap_grad_reduce
                )
            except RuntimeError:
                # gloo fallback: allreduce (shard view then holds the result)
                self.comm_handle = dist.all_reduce(
                    self.data, group=self.dp_group, async_op=self.overlap_grad_reduce
                )
        else:
            self.comm_handle = dist.all_reduce(
                self.data, group=self.dp_group, async_op=self.overlap_grad_reduce
            )

    def finish_grad_sync(self):
        if not self.comm_issued:
            self.start_grad_sync()
        if self.comm_handle is not None and self.overlap_grad_reduce:
            self.comm_handle.wait()

    def register_grad_ready(self, param):
        assert param in self.params
        self.params_with_grad.add(param)
        if (
            self.overlap_grad_reduce
            and self.is_last_microbatch
            and not self.comm_issued
            and len(self.params_with_grad) == len(self.params)
        ):
            self.start_grad_sync()


class GradBuffer:
    """One flat buffer per dtype, carved into buckets; params mapped to
    views.  Params are traversed in REVERSE module order so bucket i's
    grads complete early in backward (overlap)."""

    def __init__(
        self,
        dtype: torch.dtype,
        params: List[torch.nn.Parameter],
        dp_group,
        bucket_bytes: int,
        use_distributed_optimizer: bool,
        overlap_grad_reduce: bool,
        device: torch.device,
    ):
        self.dtype = dtype
        self.dp_group = dp_group
        self.dp_world = dist.get_world_size(dp_group) if dp_group is not None else 1
        self.use_distributed_optimizer = use_distributed_optimizer
        self.overlap_grad_reduce = overlap_grad_reduce

        align = max(1, self.dp_world) * 64
        elt = torch.finfo(dtype).bits // 8 if dtype.is_floating_point else 4
        bucket_elems = max(bucket_bytes // elt, 1)

        self.param_index_map: Dict[torch.nn.Parameter, tuple] = {}
        self.buckets: List[Bucket] = []
        self.bucket_of_param: Dict[torch.nn.Parameter, Bucket] = {}

        # assign offsets (reverse order), close bucket when > bucket_elems
        offset = 0
        cur_params: List[torch.nn.Parameter] = []
        bucket_starts: List[int] = [0]
        bucket_param_lists: List[List[torch.nn.Parameter]] = []
        for p in params[::-1]:
            n = _numel_local(p)
            self.param_index_map[p] = (offset, offset + n)
            cur_params.append(p)
            offset += n
            if offset - bucket_starts[-1] >= bucket_elems:
                offset = _alloc_aligned(offset, align)
                bucket_param_lists.append(cur_params)
                cur_params = []
                bucket_starts.append(offset)
        if cur_params:
            offset = _alloc_aligned(offset, align)
            bucket_param_lists.append(cur_params)
            bucket_starts.append(offset)
        self.numel = offset
        self.data = torch.zeros(self.numel, dtype=dtype, device=device)

        for i, plist in enumerate(bucket_param_lists):
            start, end = bucket_starts[i], bucket_starts[i + 1]
            b = Bucket(
                plist, self.data.narrow(0, start, end - start), start,
                dp_group, self.dp_world, use_distributed_optimizer,
                overlap_grad_reduce,
            )
            self.buckets.append(b)
            for p in plist:
                self.bucket_of_param[p] = b

    def get_main_grad_view(self, param) -> torch.Tensor:
        s, e = self.param_index_map[param]
        shape = (
            param.data._local_tensor.shape
            if hasattr(param.data, "_local_tensor")
            else param.data.shape
        )
        return self.data.narrow(0, s, e - s).view(shape)

    def reset(self):
        self.data.zero_()
        for b in self.buckets:
            b.reset()

    def start_grad_sync(self):
        for b in self.buckets:
            if not b.comm_issued:
                b.start_grad_sync()

    def finish_grad_sync(self):
        for b in self.buckets:
            b.finish_grad_sync()

    def register_grad_ready(self, param):
        self.bucket_of_param[param].register_grad_ready(param)

    def set_is_last_microbatch(self, flag: bool):
        for b in self.buckets:
            b.is_last_microbatch = flag


def _numel_local(p: torch.nn.Parameter) -> int:
    d = p.data
    if hasattr(d, "_local_tensor"):
        return d._local_tensor.numel()
    return d.numel()
