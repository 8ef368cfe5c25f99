"""Mesh collectives: every DTensor redistribution lowers through here to
RCCL (torch.distributed backend "nccl" on ROCm) over xGMI.

Parity target: legacy/vescale/dtensor/_collective_utils.py:50-507 (mesh_scatter,
mesh_all_to_all(_single), mesh_broadcast, mesh_reduce_scatter, mesh_all_gather,
mesh_all_reduce, broadcast_across_mesh, cost model) and the ragged scatter at
vescale/dtensor/_collective_utils.py:66 — re-designed for xGMI:

xGMI topology note (MI355X, 8-GPU node): each GPU has 7 point-to-point
Infinity Fabric links at ~153 GB/s — fully connected, NO switch.  Ring
algorithms are bounded by ONE link (~153 GB/s); RCCL's direct/one-shot
algorithms use all 7 links (~1 TB/s aggregate).  The cost model below is
parameterized for that (vs the reference's NVLink-aggregate assumptions),
and the ragged scatter is a single uneven all-to-all rather than the
reference's serialized send/recv loop (its own TODO flags that perf bug).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from .device_mesh import DeviceMesh


def _supports_a2a(pg) -> bool:
    try:
        return dist.get_backend(pg) != "gloo"
    except Exception:
        return True


def _get_op(reduce_op: str) -> dist.ReduceOp:
    return {
        "sum": dist.ReduceOp.SUM,
        "avg": dist.ReduceOp.AVG if hasattr(dist.ReduceOp, "AVG") else dist.ReduceOp.SUM,
        "max": dist.ReduceOp.MAX,
        "min": dist.ReduceOp.MIN,
        "product": dist.ReduceOp.PRODUCT,
    }[reduce_op]


def _is_meta(t: torch.Tensor) -> bool:
    """Meta tensors short-circuit every collective: c10d has no meta
    kernels, and meta-device DTensors exist exactly to propagate shapes
    and placements with zero comms (reference test/dtensor/meta_device)."""
    return t.device.type == "meta"


def mesh_all_reduce(
    tensor: torch.Tensor, mesh: DeviceMesh, reduce_op: str = "sum", mesh_dim: int = 0,
    async_op: bool = False,
):
    if mesh.size(mesh_dim) == 1 or _is_meta(tensor):
        return None
    pg = mesh.get_group(mesh_dim)
    if reduce_op == "avg" and not hasattr(dist.ReduceOp, "AVG"):
        work = dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=pg, async_op=async_op)
        tensor.div_(mesh.size(mesh_dim))
        return work
    return dist.all_reduce(tensor, op=_get_op(reduce_op), group=pg, async_op=async_op)


def mesh_broadcast(
    tensor: torch.Tensor, mesh: DeviceMesh, mesh_dim: int = 0, src_mesh_rank: int = 0,
    async_op: bool = False,
):
    if mesh.size(mesh_dim) == 1 or _is_meta(tensor):
        return None
    pg = mesh.get_group(mesh_dim)
    src_global = dist.get_global_rank(pg, src_mesh_rank)
    return dist.broadcast(tensor, src=src_global, group=pg, async_op=async_op)


def mesh_all_gather(
    tensor: torch.Tensor,
    mesh: DeviceMesh,
    mesh_dim: int = 0,
    *,
    gather_dim: int = 0,
    out: Optional[torch.Tensor] = None,
    sizes: Optional[Sequence[int]] = None,
    async_op: bool = False,
):
    """All-gather along mesh_dim.  Even case uses all_gather_into_tensor
    (single fused RCCL call); uneven (`sizes` given, per-rank sizes on
    gather_dim) uses list all_gather then narrow+cat."""
    ws = mesh.size(mesh_dim)
    if ws == 1:
        return tensor
    if _is_meta(tensor):
        total = sum(sizes) if sizes is not None else ws * tensor.shape[gather_dim]
        shape = list(tensor.shape)
        shape[gather_dim] = total
        out = tensor.new_empty(shape)
        return (out, None) if async_op else out
    pg = mesh.get_group(mesh_dim)
    if sizes is None:
        t = tensor.contiguous()
        if gather_dim != 0:
            t = t.movedim(gather_dim, 0).contiguous()
        if out is None:
            out = t.new_empty((ws * t.shape[0],) + tuple(t.shape[1:]))
        work = dist.all_gather_into_tensor(out, t, group=pg, async_op=async_op)
        if async_op:
            return out, work
        if gather_dim != 0:
            out = out.reshape((ws,) + tuple(t.shape)).movedim(1, gather_dim + 1)
            out = out.reshape(
                tuple(tensor.shape[:gather_dim])
                + (ws * tensor.shape[gather_dim],)
                + tuple(tensor.shape[gather_dim + 1 :])
            ) if False else torch.cat(list(out.unbind(0)), dim=gather_dim)
        return out
    # uneven path (ragged unshard / uneven Shard): pad-free list gather
    max_size = max(sizes)
    t = tensor.movedim(gather_dim, 0).contiguous() if gather_dim != 0 else tensor.contiguous()
    rest = tuple(t.shape[1:])
    padded = t.new_empty((max_size,) + rest)
    padded[: t.shape[0]] = t
    gathered = [t.new_empty((max_size,) + rest) for _ in range(ws)]
    dist.all_gather(gathered, padded, group=pg)
    pieces = [g.narrow(0, 0, s) for g, s in zip(gathered, sizes)]
    out = torch.cat(pieces, dim=0)
    if gather_dim != 0:
        out = out.movedim(0, gather_dim).contiguous()
    return out


def mesh_reduce_scatter(
    tensor: torch.Tensor,
    mesh: DeviceMesh,
    reduce_op: str = "sum",
    mesh_dim: int = 0,
    *,
    scatter_dim: int = 0,
    async_op: bool = False,
):
    """Reduce-scatter: tensor's scatter_dim must be divisible by mesh size
    (uneven callers pad first in redistribute)."""
    ws = mesh.size(mesh_dim)
    if ws == 1:
        return tensor
    if _is_meta(tensor):
        shape = list(tensor.shape)
        assert shape[scatter_dim] % ws == 0
        shape[scatter_dim] //= ws
        out = tensor.new_empty(shape)
        return (out, None) if async_op else out
    pg = mesh.get_group(mesh_dim)
    t = tensor.movedim(scatter_dim, 0).contiguous() if scatter_dim != 0 else tensor.contiguous()
    assert t.shape[0] % ws == 0, f"reduce_scatter dim {t.shape[0]} % {ws} != 0"
    out_shape = (t.shape[0] // ws,) + tuple(t.shape[1:])
    out = t.new_empty(out_shape)
    op = _get_op(reduce_op if reduce_op != "avg" else "sum")
    work = dist.reduce_scatter_tensor(out, t, op=op, group=pg, async_op=async_op)
    if reduce_op == "avg":
        out.div_(ws)
    if async_op:
        return out, work
    if scatter_dim != 0:
        out = out.movedim(0, scatter_dim).contiguous()
    return out


def mesh_scatter(
    output: torch.Tensor,
    scatter_list: Optional[List[torch.Tensor]],
    mesh: DeviceMesh,
    mesh_dim: int = 0,
    src_mesh_rank: int = 0,
    async_op: bool = False,
):
    if _is_meta(output):
        return None
    pg = mesh.get_group(mesh_dim)
    src_global = dist.get_global_rank(pg, src_mesh_rank)
    if dist.get_rank() == src_global:
        return dist.scatter(output, scatter_list, src=src_global, group=pg, async_op=async_op)
    return dist.scatter(output, None, src=src_global, group=pg, async_op=async_op)


def mesh_scatter_ragged(
    output: torch.Tensor,
    full_flat: Optional[torch.Tensor],
    split_sizes: Sequence[int],
    mesh: DeviceMesh,
    mesh_dim: int = 0,
    src_mesh_rank: int = 0,
):
    """Uneven scatter of a flat tensor — used to distribute a RaggedShard.

    The reference serializes send/recv from the root (its noted perf bug at
    vescale/dtensor/_collective_utils.py:65,83).  Here: a single uneven
    all_to_all_single where only the root contributes input — one RCCL call,
    and on xGMI the root's 7 links fan out concurrently.
    """
    if _is_meta(output):
        return output
    pg = mesh.get_group(mesh_dim)
    ws = mesh.size(mesh_dim)
    my = mesh.get_local_rank(mesh_dim) if mesh.get_coordinate() is not None else -1
    in_splits = [0] * ws
    if my == src_mesh_rank:
        assert full_flat is not None
        in_splits = list(split_sizes)
        inp = full_flat.contiguous()
    else:
        inp = output.new_empty(0)
    out_splits = [0] * ws
    out_splits[src_mesh_rank] = output.numel()
    dist.all_to_all_single(
        output.view(-1), inp.view(-1), out_splits, in_splits, group=pg
    )
    return output


def mesh_all_to_all_single(
    output: torch.Tensor,
    input: torch.Tensor,
    mesh: DeviceMesh,
    mesh_dim: int = 0,
    *,
    output_split_sizes: Optional[Sequence[int]] = None,
    input_split_sizes: Optional[Sequence[int]] = None,
    async_op: bool = False,
):
    ws = mesh.size(mesh_dim)
    if ws == 1:
        output.copy_(input.view_as(output))
        return None
    if _is_meta(output):
        return None
    pg = mesh.get_group(mesh_dim)
    if _supports_a2a(pg):
        return dist.all_to_all_single(
            output, input,
            list(output_split_sizes) if output_split_sizes is not None else None,
            list(input_split_sizes) if input_split_sizes is not None else None,
            group=pg, async_op=async_op,
        )
    # gloo fallback: decompose into isend/irecv pairs
    my = dist.get_rank(pg)
    iss = list(input_split_sizes) if input_split_sizes is not None else [input.numel() // ws] * ws
    oss = list(output_split_sizes) if output_split_sizes is not None else [output.numel() // ws] * ws
    in_chunks, off = [], 0
    flat_in = input.contiguous().view(-1)
    for s in iss:
        in_chunks.append(flat_in.narrow(0, off, s))
        off += s
    out_chunks, off = [], 0
    flat_out = output.view(-1)
    for s in oss:
        out_chunks.append(flat_out.narrow(0, off, s))
        off += s
    _p2p_exchange(out_chunks, in_chunks, pg, my, ws)
    return None


def mesh_all_to_all(
    output_list: List[torch.Tensor],
    input_list: List[torch.Tensor],
    mesh: DeviceMesh,
    mesh_dim: int = 0,
    async_op: bool = False,
):
    ws = mesh.size(mesh_dim)
    if ws == 1:
        output_list[0].copy_(input_list[0])
        return None
    if output_list and _is_meta(output_list[0]):
        return None
    pg = mesh.get_group(mesh_dim)
    if _supports_a2a(pg):
        return dist.all_to_all(output_list, input_list, group=pg, async_op=async_op)
    my = dist.get_rank(pg)
    send = [t.contiguous() for t in input_list]
    recv_bufs = [t if t.is_contiguous() else t.contiguous() for t in output_list]
    _p2p_exchange(recv_bufs, send, pg, my, ws)
    for o, b in zip(output_list, recv_bufs):
        if o.data_ptr() != b.data_ptr():
            o.copy_(b)
    return None


def _p2p_exchange(recv_chunks, send_chunks, pg, my, ws):
    """Pairwise deadlock-free exchange (gloo fallback for all_to_all)."""
    recv_chunks[my].copy_(send_chunks[my].view_as(recv_chunks[my]))
    reqs = []
    for peer in range(ws):
        if peer == my:
            continue
        g = dist.get_global_rank(pg, peer)
        if recv_chunks[peer].numel():
            reqs.append(dist.irecv(recv_chunks[peer], src=g, group=pg))
        if send_chunks[peer].numel():
            reqs.append(dist.isend(send_chunks[peer], dst=g, group=pg))
    for r in reqs:
        r.wait()


def broadcast_across_mesh(
    tensor: torch.Tensor, src_rank: int, dst_ranks: Sequence[int], tag: int = 0
):
    """P2P broadcast from a rank in one mesh to ranks of another (PP spec
    exchange).  Uses the default PG."""
    if _is_meta(tensor):
        return tensor
    me = dist.get_rank()
    reqs = []
    if me == src_rank:
        for d in dst_ranks:
            if d == me:
                continue
            reqs.append(dist.isend(tensor, dst=d, tag=tag))
    elif me in dst_ranks:
        reqs.append(dist.irecv(tensor, src=src_rank, tag=tag))
    for r in reqs:
        r.wait()
    return tensor


# ---------------------------------------------------------------------------
# α-β cost model, xGMI-tuned.
#
# MI355X node: fully-connected 8 GPUs, 7 links × ~153 GB/s unidirectional
# per GPU.  For a collective over w ranks within one node:
#   - ring algorithms move (w-1)/w of the data over ONE link  -> B_ring = 153
#   - direct (one-shot) uses min(w-1,7) links concurrently    -> B_dir = 153*(w-1)
# RCCL picks per-size; we model the envelope.  Latency α ≈ 8 µs per hop
# intra-node.  Inter-node (IB) modeled at 50 GB/s/GPU.
# Costs are returned in microseconds (same convention as the reference model
# at legacy/vescale/dtensor/_collective_utils.py:411-470).
# ---------------------------------------------------------------------------
XGMI_LINK_GBPS = 153.0
XGMI_NUM_LINKS = 7
IB_GBPS = 50.0
ALPHA_US = 8.0


def _bw_bytes_per_us(num_ranks: int, intra_node: bool = True) -> float:
    if not intra_node:
        return IB_GBPS * 1e3 / 1e6 * 1e3  # GB/s -> bytes/us : 50e9/1e6 = 5e4
    links = min(num_ranks - 1, XGMI_NUM_LINKS)
    return XGMI_LINK_GBPS * max(1, links) * 1e3  # bytes/us (153e9/1e6 = 153e3 per link)


def allgather_cost(bytes_gb: float, num_ranks: int) -> float:
    if num_ranks <= 1:
        return 0.0
    nbytes = bytes_gb * 1e9
    per_rank = nbytes / num_ranks
    bw = _bw_bytes_per_us(num_ranks)
    return ALPHA_US * (num_ranks - 1) / max(1, num_ranks - 1) + per_rank * (num_ranks - 1) / bw


def reduce_scatter_cost(bytes_gb: float, num_ranks: int) -> float:
    return allgather_cost(bytes_gb, num_ranks)


def allreduce_cost(bytes_gb: float, num_ranks: int) -> float:
    return allgather_cost(bytes_gb, num_ranks) * 2.0


def alltoall_cost(bytes_gb: float, num_ranks: int) -> float:
    if num_ranks <= 1:
        return 0.0
    nbytes = bytes_gb * 1e9
    bw = _bw_bytes_per_us(num_ranks)
    return ALPHA_US + nbytes * (num_ranks - 1) / num_ranks / bw


def redistribute_cost(src_spec, dst_spec) -> float:
    """Sum of per-mesh-dim transition costs; used by sharding-prop strategy
    selection to pick the cheapest input placements."""
    from .placement_types import Partial, Replicate, Shard

    if src_spec.mesh != dst_spec.mesh:
        return float("inf")
    cost = 0.0
    gb = src_spec.bytes() / 1e9
    for dim, (s, d) in enumerate(zip(src_spec.placements, dst_spec.placements)):
        w = src_spec.mesh.size(dim)
        if s == d:
            continue
        if isinstance(s, Shard) and isinstance(d, Replicate):
            cost += allgather_cost(gb, w)
        elif isinstance(s, Partial) and isinstance(d, Replicate):
            cost += allreduce_cost(gb, w)
        elif isinstance(s, Partial) and isinstance(d, Shard):
            cost += reduce_scatter_cost(gb, w)
        elif isinstance(s, Shard) and isinstance(d, Shard):
            cost += alltoall_cost(gb, w)
        elif isinstance(s, Replicate):
            cost += 0.0  # local slice
        else:
            cost += allreduce_cost(gb, w)
    return cost


def wait(t):
    """Wait on an async-collective tensor/work pair."""
    if isinstance(t, tuple):
        tensor, work = t
        if work is not None:
            work.wait()
        return tensor
    return t
