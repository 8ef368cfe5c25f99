"""Per-parameter, global-index-space optimizer-state checkpointing.

Parity: the reference's DistributedOptimizer flat-range checkpointing
(`optim/distributed_optimizer.py:51` OptimizerStateSpec + the planner's
`collect_optim_state_across_dp_ranks`) — each rank's slice of the flat
fp32 optimizer state is addressed as axis-aligned BOXES in the owning
PARAMETER'S UNSHARDED index space, so a saved checkpoint can be reloaded
at a different data-parallel size AND a different tensor-parallel size
(the flat ranges re-segment; the boxes don't care).

A parameter that is tensor-parallel sharded carries `_tp_shard =
(dim, global_dim_size, [(local_start, n, global_start), ...])` metadata
(segments along `dim` — the fused/packed layouts like wqkv or w13 have
several segments).  See models/llama.py for the annotations.
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import torch

from .ragged_boxes import Box, box_numel, break_ragged_box

__all__ = ["FlatParamStateShard", "build_param_state_entries"]


def _segments_for(
    param: torch.Tensor, local_shape: Sequence[int]
) -> Tuple[int, int, List[Tuple[int, int, int]]]:
    """(tp_dim, global_dim_size, segments) — identity if not TP-sharded.
    local_shape comes from the unit's param_infos: the LIVE param may be
    freed (FSDP resharded) with an empty shape."""
    meta = getattr(param, "_tp_shard", None)
    if meta is None:
        d0 = local_shape[0] if len(local_shape) else 1
        return 0, d0, [(0, d0, 0)]
    return meta


def _global_boxes(
    local_shape: Sequence[int], a: int, b: int, param: torch.Tensor
) -> List[Tuple[Box, int]]:
    """Boxes (in the param's GLOBAL shape) covering the LOCAL flat range
    [a, b), paired with the local flat start of each box."""
    local_shape = tuple(local_shape)
    dim, gdim, segs = _segments_for(param, local_shape)
    out: List[Tuple[Box, int]] = []
    if dim == 0:
        inner = 1
        for s in local_shape[1:]:
            inner *= s
        for l0, n, g0 in segs:
            lo = max(a, l0 * inner)
            hi = min(b, (l0 + n) * inner)
            if lo >= hi:
                continue
            for off, sz in break_ragged_box(local_shape, lo, hi):
                goff = (off[0] - l0 + g0,) + tuple(off[1:])
                # local flat start of THIS box:
                fs = 0
                stride = 1
                for o, s in zip(reversed(off), reversed(local_shape)):
                    fs += o * stride
                    stride *= s
                out.append(((goff, tuple(sz)), fs))
        return out
    # single-segment shard along dim k > 0
    assert len(segs) == 1, "multi-segment TP supported on dim 0 only"
    l0, n, g0 = segs[0]
    for off, sz in break_ragged_box(local_shape, a, b):
        goff = tuple(
            o + (g0 - l0 if d == dim else 0) for d, o in enumerate(off)
        )
        fs = 0
        stride = 1
        for o, s in zip(reversed(off), reversed(local_shape)):
            fs += o * stride
            stride *= s
        out.append(((goff, tuple(sz)), fs))
    return out


class FlatParamStateShard(torch.Tensor):
    """One parameter's slice of a rank's flat optimizer-state shard,
    checkpoint-addressed in the parameter's GLOBAL index space.  A tensor
    wrapper-subclass (torch DCP's find_state_dict_object requires a
    torch.Tensor) exposing the `_Checkpointable` dunders, like our DTensor
    (dtensor_dcp.py) — which makes the saved state reloadable at any
    DP x TP topology."""

    @staticmethod
    def __new__(
        cls,
        local_flat: torch.Tensor,           # this rank's slice (1-D view)
        global_shape: Sequence[int],
        boxes: List[Tuple[Box, int]],       # (global box, local flat start)
        base: int = 0,                      # param-local flat start of _local
    ):
        r = torch.Tensor._make_wrapper_subclass(
            cls,
            torch.Size(global_shape),
            dtype=local_flat.dtype,
            device=local_flat.device,
            requires_grad=False,
        )
        r._local = local_flat
        r._boxes = boxes
        r._flat_base = base
        return r

    def __repr__(self):  # noqa: D105
        return f"FlatParamStateShard(shape={tuple(self.shape)}, boxes={len(self._boxes)})"

    @classmethod
    def __torch_dispatch__(cls, func, types, args=(), kwargs=None):
        # checkpoint-only object: allow the trivial ops DCP may issue
        name = str(func)
        if "detach" in name or "clone" in name or "alias" in name:
            return args[0]
        raise NotImplementedError(
            f"FlatParamStateShard is a checkpoint container; op {func} unsupported"
        )

    # --- _Checkpointable protocol (see checkpoint/dtensor_dcp.py) ---
    def __create_write_items__(self, fqn: str, object) -> list:
        from torch.distributed.checkpoint.metadata import (
            ChunkStorageMetadata,
            MetadataIndex,
            TensorProperties,
        )
        from torch.distributed.checkpoint.planner import (
            TensorWriteData,
            WriteItem,
            WriteItemType,
        )

        out = []
        for (off, sz), _fs in self._boxes:
            out.append(
                WriteItem(
                    index=MetadataIndex(fqn, torch.Size(off)),
                    type=WriteItemType.SHARD,
                    tensor_data=TensorWriteData(
                        chunk=ChunkStorageMetadata(
                            offsets=torch.Size(off), sizes=torch.Size(sz)
                        ),
                        properties=TensorProperties(dtype=self.dtype),
                        size=self.shape,
                    ),
                )
            )
        return out

    def __create_chunk_list__(self) -> list:
        from torch.distributed.checkpoint.metadata import ChunkStorageMetadata

        return [
            ChunkStorageMetadata(offsets=torch.Size(off), sizes=torch.Size(sz))
            for (off, sz), _fs in self._boxes
        ]

    def __get_tensor_shard__(self, index) -> torch.Tensor:
        # box flat starts are relative to the PARAM's local tensor;
        # self._local starts at the first covered element (_flat_base)
        for (off, sz), fs in self._boxes:
            if torch.Size(off) == index.offset:
                return self._local.narrow(
                    0, fs - self._flat_base, box_numel((off, sz))
                ).view(sz)
        raise ValueError(f"no state box at {index.offset} for {index.fqn}")


def build_param_state_entries(
    unit,
    state: Dict[str, torch.Tensor],
    prefix: str,
) -> Dict[str, object]:
    """Map a unit's flat state tensors to per-param global-space entries.

    unit: fsdp FSDPUnit (param_infos + shard_off/shard_numel + params).
    state: {"m": flat_shard, "v": ..., "master": ...} (1-D, shard_numel).
    Returns {f"{prefix}.{param_fqn}.{key}": FlatParamStateShard | ...}.
    """
    out: Dict[str, object] = {}
    s0 = unit.shard_off
    s1 = s0 + unit.shard_numel
    for (fqn, shape, p_off, p_numel), p in zip(unit.param_infos, unit.params):
        lo = max(s0, p_off)
        hi = min(s1, p_off + p_numel)
        if lo >= hi:
            continue
        a = lo - p_off              # param-local flat range
        b = hi - p_off
        boxes = _global_boxes(shape, a, b, p)
        gshape = _global_shape_for(p, shape)
        for key, t in state.items():
            if not torch.is_tensor(t) or t.ndim != 1:
                continue
            seg = t.narrow(0, lo - s0, hi - lo)
            out[f"{prefix}.{fqn}.{key}"] = FlatParamStateShard(
                seg, gshape, boxes, base=a
            )
    return out


def _global_shape_for(p: torch.Tensor, local_shape) -> Tuple[int, ...]:
    meta = getattr(p, "_tp_shard", None)
    shape = list(local_shape)
    if meta is not None:
        dim, gdim, _ = meta
        shape[dim] = gdim
    return tuple(shape)
