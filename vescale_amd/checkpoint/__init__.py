"""vescale_amd.checkpoint — distributed checkpointing in the standard DCP
on-disk format.

Parity: legacy/vescale/checkpoint/__init__.py:16-60 (save/load API),
api/vescale_checkpointer.py (async futures, broadcast-load), with our
DTensor plugged into DCP through the _Checkpointable dunders
(dtensor_dcp.py), plan caching + balanced dedup (planner.py), and the
pinned D2H pool (mem_pool.py).  Load-time RESHARDING across different
(dp, tp, pp) layouts is inherited from DCP chunk resolution over our
chunk decompositions (incl. ragged flat ranges as N-D boxes).
"""
from __future__ import annotations

import concurrent.futures
import logging
import os
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist
import torch.distributed.checkpoint as dcp

from . import dtensor_dcp  # installs DTensor DCP hooks  # noqa: F401
from .mem_pool import (  # noqa: F401
    mem_checkpoint_path,
    GLOBAL_POOL,
    PinnedStoragePool,
    copy_gpu_tensor_to_cpu_pinned_mem_pool,
    release_cpu_tensor,
)
from .mem_server import MemCheckpointServer, fetch_checkpoint  # noqa: F401
from .planner import VeScaleLoadPlanner, VeScaleSavePlanner
from .ragged_boxes import break_ragged_box

__all__ = [
    "VeScaleCheckpointer",
    "CheckpointState",
    "save",
    "load",
    "VeScaleSavePlanner",
    "VeScaleLoadPlanner",
    "PinnedStoragePool",
    "MemCheckpointServer",
    "fetch_checkpoint",
    "copy_gpu_tensor_to_cpu_pinned_mem_pool",
    "break_ragged_box",
]

_executor = concurrent.futures.ThreadPoolExecutor(max_workers=2)
_pending: list = []


def _materialize_state(obj):
    """Resolve .state_dict()-bearing objects (models/optimizers/engines)."""
    if hasattr(obj, "sharded_state_dict"):
        return obj.sharded_state_dict()
    if hasattr(obj, "state_dict"):
        return obj.state_dict()
    return obj


logger = logging.getLogger("vescale_amd.checkpoint")
if os.environ.get("VESCALE_CHECKPOINT_LOGGING_LEVEL"):
    # reference env flag: set the checkpoint subsystem's log level by name
    logger.setLevel(os.environ["VESCALE_CHECKPOINT_LOGGING_LEVEL"].upper())


def save(
    path: str,
    checkpoint_state: Dict[str, Any],
    *,
    async_checkpoint: bool = False,
    process_group=None,
):
    """checkpoint_state: {"model": module_or_state_dict, "optimizer": ...}.
    Each component is written to its own DCP directory under `path`."""
    futures = []
    for key, obj in checkpoint_state.items():
        sd = _materialize_state(obj)
        comp_path = os.path.join(path, key)
        if dist.is_initialized() and dist.get_rank() == 0:
            os.makedirs(comp_path, exist_ok=True)
        elif not dist.is_initialized():
            os.makedirs(comp_path, exist_ok=True)
        if dist.is_initialized():
            dist.barrier()
        logger.debug("checkpoint save: component %s -> %s", key, comp_path)
        if async_checkpoint:
            if dist.is_initialized():
                # dcp.async_save stages (D2H) synchronously, then the
                # serialization + write run in a background thread with the
                # plan collectives coordinated safely (reference async
                # futures, api/vescale_checkpointer.py:85)
                fut = dcp.async_save(
                    sd,
                    storage_writer=dcp.FileSystemWriter(comp_path),
                    planner=VeScaleSavePlanner(),
                    process_group=process_group,
                )
            else:
                # single process: stage to CPU (pinned pool on GPU), write in
                # a background thread
                cpu_sd = _to_cpu(sd)
                fut = _executor.submit(_do_save, comp_path, cpu_sd, process_group)
            futures.append(fut)
            _pending.append(fut)
        else:
            _do_save(comp_path, sd, process_group)
    return futures if async_checkpoint else None


def _to_cpu(sd):
    out = {}
    for k, v in sd.items():
        if isinstance(v, torch.Tensor) and not hasattr(v, "_spec") and v.is_cuda:
            out[k] = copy_gpu_tensor_to_cpu_pinned_mem_pool(v)
        else:
            out[k] = v
    return out


def _do_save(path, sd, process_group):
    dcp.save(
        sd,
        storage_writer=dcp.FileSystemWriter(path),
        planner=VeScaleSavePlanner(),
        process_group=process_group,
    )


def wait_pending():
    """Block until all async checkpoints finish (reference: futures on the
    VeScaleCheckpointer)."""
    for f in list(_pending):
        f.result()
    _pending.clear()


def _iter_state_tensors(obj):
    """Deterministic-order walk over every tensor leaf of a state dict
    (sorted keys, so all ranks traverse identically)."""
    if isinstance(obj, dict):
        for k in sorted(obj.keys(), key=str):
            yield from _iter_state_tensors(obj[k])
    elif isinstance(obj, (list, tuple)):
        for v in obj:
            yield from _iter_state_tensors(v)
    elif isinstance(obj, torch.Tensor):
        yield obj


def _broadcast_state(sd, group) -> None:
    """Broadcast every tensor leaf in-place from group-rank 0."""
    src = dist.get_global_rank(group, 0) if group is not None else 0
    for t in _iter_state_tensors(sd):
        local = t._local_tensor if hasattr(t, "_local_tensor") else t
        dist.broadcast(local, src=src, group=group)


def load(
    path: str,
    checkpoint_state: Dict[str, Any],
    *,
    broadcast_checkpoint: bool = False,
    process_group=None,
):
    """In-place load into the given model/optimizer objects (resharding as
    needed).

    broadcast_checkpoint: only rank 0 of `process_group` (default: WORLD)
    reads from storage (single-rank DCP read of its local plan) and every
    tensor leaf is broadcast in-place to the other ranks (reference
    api/vescale_checkpointer.py:160-214 DP-broadcast load).  Contract: the
    state must be REPLICATED across the group (DDP replicas over DP) — for
    DP-sharded state (FSDP/RaggedShard over the same group) use the default
    per-rank DCP read, which fetches exactly each rank's chunks.
    """
    use_bcast = (
        broadcast_checkpoint
        and dist.is_initialized()
        and dist.get_world_size(process_group) > 1
    )
    group_rank = dist.get_rank(process_group) if use_bcast else 0
    for key, obj in checkpoint_state.items():
        comp_path = os.path.join(path, key)
        sd = _materialize_state(obj)
        if use_bcast:
            if group_rank == 0:
                dcp.load(
                    sd,
                    storage_reader=dcp.FileSystemReader(comp_path),
                    planner=VeScaleLoadPlanner(),
                    no_dist=True,
                )
            _broadcast_state(sd, process_group)
        else:
            dcp.load(
                sd,
                storage_reader=dcp.FileSystemReader(comp_path),
                planner=VeScaleLoadPlanner(),
                process_group=process_group,
            )
        # push back into stateful objects that need it
        if hasattr(obj, "load_sharded_state_dict"):
            obj.load_sharded_state_dict(sd)
        elif hasattr(obj, "load_state_dict") and not hasattr(obj, "sharded_state_dict"):
            try:
                obj.load_state_dict(sd)
            except Exception:
                pass  # in-place DCP load already mutated the tensors
    return checkpoint_state


# reference-shape surface (api/vescale_checkpointer.py): a class namespace
# over the same save/load; CheckpointState is the {"model": ..., ...} dict
CheckpointState = Dict[str, Any]


class VeScaleCheckpointer:
    """Classmethod namespace over save/load (reference VeScaleCheckpointer)."""

    @classmethod
    def save(cls, path: str, checkpoint_state: CheckpointState,
             async_checkpoint: bool = False):
        return save(path, checkpoint_state, async_checkpoint=async_checkpoint)

    @classmethod
    def load(cls, path: str, checkpoint_state: CheckpointState,
             broadcast_checkpoint: bool = False):
        return load(path, checkpoint_state,
                    broadcast_checkpoint=broadcast_checkpoint)
