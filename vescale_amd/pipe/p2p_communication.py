"""Pipeline p2p layer over RCCL/xGMI (intra-node) or IB (inter-node).

Parity: legacy/vescale/pipe/p2p_communication.py:60-1005 — dynamic-shape
handshake (:125 _communicate_shapes), batched isend/irecv
(:219 _batched_p2p_ops), ordered fallback, and the public
send_forward / recv_forward / send_backward / recv_backward /
send_forward_recv_backward / send_backward_recv_forward API.

Deadlock rule on RCCL: both peers must issue their op lists in matching
order — batch_isend_irecv guarantees this by fusing into one RCCL group.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

_MAX_DIMS = 8

_DTYPES = [
    torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int32,
    torch.uint8, torch.bool, torch.float64,
]


def _use_batched(pg) -> bool:
    try:
        return dist.get_backend(pg) != "gloo"
    except Exception:
        return False


def _p2p_device(pg) -> torch.device:
    """Device for p2p payloads AND meta tensors.  RCCL (backend 'nccl' on
    ROCm) rejects CPU tensors in batch_isend_irecv, so the shape handshake
    must ride device tensors on NCCL groups (reference _communicate_shapes
    allocates on the compute device, p2p_communication.py:125)."""
    try:
        backend = str(dist.get_backend(pg))
    except Exception:
        backend = "gloo"
    if "nccl" in backend and torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _meta_encode(t: torch.Tensor, device=None) -> torch.Tensor:
    m = torch.zeros(_MAX_DIMS + 2, dtype=torch.int64)
    m[0] = t.ndim
    for i, s in enumerate(t.shape):
        m[1 + i] = s
    m[_MAX_DIMS + 1] = _DTYPES.index(t.dtype)
    return m.to(device) if device is not None else m


def _meta_decode(m: torch.Tensor) -> Tuple[Tuple[int, ...], torch.dtype]:
    m = m.cpu()
    nd = int(m[0])
    shape = tuple(int(m[1 + i]) for i in range(nd))
    return shape, _DTYPES[int(m[_MAX_DIMS + 1])]


# Shape cache: after the first handshake for a (pg, peer, direction) key the
# metadata is reused, skipping the extra p2p round-trip for every microbatch
# (reference REUSE_COMM_SHAPE env, pipe/p2p_communication.py:125).  Enabled by
# default; shapes that change per-microbatch need VESCALE_REUSE_COMM_SHAPE=0.
_SHAPE_CACHE: dict = {}


def _reuse_shapes() -> bool:
    import os

    return os.environ.get("VESCALE_REUSE_COMM_SHAPE", "1") not in ("0", "false")


def reset_shape_cache():
    _SHAPE_CACHE.clear()


def _communicate_shapes(send_t: Optional[torch.Tensor], recv_from: Optional[int],
                        send_to: Optional[int], pg) -> Optional[Tuple]:
    """Exchange (shape, dtype) metadata before variable-shape transfers
    (reference :125).

    Symmetry invariant for the reuse cache: once a (pg, peer, direction)
    channel has handshaked ONCE, both the sender and the receiver skip all
    later handshakes on that channel unconditionally — a one-sided skip
    would orphan a meta send into the peer's payload irecv.  Hence shapes
    must be static per channel while VESCALE_REUSE_COMM_SHAPE=1 (default;
    the pipeline emitter sends fixed-shape microbatches).
    """
    reuse = _reuse_shapes()
    rkey = (id(pg), "recv", recv_from)
    skey = (id(pg), "send", send_to)
    do_recv = recv_from is not None and not (reuse and rkey in _SHAPE_CACHE)
    do_send = (
        send_to is not None and send_t is not None
        and not (reuse and skey in _SHAPE_CACHE)
    )
    dev = _p2p_device(pg)
    ops = []
    recv_buf = None
    if do_recv:
        recv_buf = torch.zeros(_MAX_DIMS + 2, dtype=torch.int64, device=dev)
        ops.append(dist.P2POp(dist.irecv, recv_buf, peer=recv_from, group=pg))
    if do_send:
        ops.append(dist.P2POp(
            dist.isend, _meta_encode(send_t, dev), peer=send_to, group=pg))
        if reuse:
            _SHAPE_CACHE[skey] = (tuple(send_t.shape), send_t.dtype)
    if ops:
        _run_p2p_ops(ops, pg)
        # NOTE: no drain here.  Draining inside the handshake turns an async
        # send stream into a sync point and deadlocks dependency-valid
        # schedules whose two ranks sit at different instructions (seen with
        # ZB-V: rank A drains its queued payload sends while rank B waits
        # for A's next meta).  Meta buffers stay alive on the send queue
        # until the end-of-schedule drain like any payload.
    if recv_buf is not None:
        meta = _meta_decode(recv_buf)
        if reuse:
            _SHAPE_CACHE[rkey] = meta
        return meta
    if recv_from is not None:
        return _SHAPE_CACHE.get(rkey)
    return None


# Outstanding async sends: (work, tensor) — the tensor reference keeps the
# buffer alive until the receiver lands (reference p2p_communication.py:71
# drain_send_reqs).  Sends NEVER block the instruction stream; recvs do.
_SEND_QUEUE: List[Tuple] = []


def drain_send_reqs():
    for work, _buf in _SEND_QUEUE:
        work.wait()
    _SEND_QUEUE.clear()


def _log_p2p(ops) -> None:
    """VESCALE_DUMMY_P2P / debug tracing (reference dtensor/_diff.py:32
    dummy_p2p): when VESCALE_DUMMY_P2P is set, log every p2p batch through
    DebugLogger (per-rank files / stderr) — schedule-debugging visibility
    without reading tensors."""
    import os

    if not os.environ.get("VESCALE_DUMMY_P2P"):
        return
    import logging

    try:
        desc = ", ".join(
            f"{'send' if o.op is dist.isend else 'recv'}"
            f"(peer={o.peer}, shape={tuple(o.tensor.shape)})"
            for o in ops
        )
    except Exception:
        desc = f"{len(ops)} ops"
    rank = dist.get_rank() if dist.is_initialized() else 0
    logging.getLogger("vescale_amd.pipe.p2p").warning(
        "[rank %d] p2p: %s", rank, desc
    )


def _run_p2p_ops(ops: List[dist.P2POp], pg):
    _log_p2p(ops)
    """Post all ops; WAIT only on receives, queue sends for a later drain.
    Pre-posting-free deadlock safety: a send can then never participate in
    a rendezvous cycle on the host side."""
    if not ops:
        return
    if _use_batched(pg):
        reqs = dist.batch_isend_irecv(ops)
        if len(reqs) == len(ops):
            for op, r in zip(ops, reqs):
                if op.op is dist.irecv:
                    r.wait()
                else:
                    _SEND_QUEUE.append((r, op.tensor))
        else:
            # coalesced group: a single work; NCCL wait() is stream-side
            for r in reqs:
                r.wait()
        return
    recv_reqs = []
    for op in sorted(ops, key=lambda o: (o.peer, 0 if o.op is dist.irecv else 1)):
        if op.op is dist.irecv:
            recv_reqs.append(dist.irecv(op.tensor, src=op.peer, group=op.group))
        else:
            _SEND_QUEUE.append(
                (dist.isend(op.tensor, dst=op.peer, group=op.group), op.tensor)
            )
    for r in recv_reqs:
        r.wait()


def _communicate(
    *,
    tensor_send_prev: Optional[torch.Tensor] = None,
    tensor_send_next: Optional[torch.Tensor] = None,
    recv_prev: bool = False,
    recv_next: bool = False,
    prev_rank: Optional[int] = None,
    next_rank: Optional[int] = None,
    pg=None,
    recv_shape=None,
    recv_dtype=None,
    device=None,
):
    """Core combined exchange (reference :411)."""
    tensor_recv_prev = tensor_recv_next = None
    # shape handshakes where needed
    if recv_prev and recv_shape is None:
        meta = _communicate_shapes(tensor_send_next, prev_rank, next_rank, pg)
        if meta:
            recv_shape, recv_dtype = meta
    elif recv_next and recv_shape is None:
        meta = _communicate_shapes(tensor_send_prev, next_rank, prev_rank, pg)
        if meta:
            recv_shape, recv_dtype = meta
    elif tensor_send_next is not None and not recv_prev and prev_rank is None:
        pass

    dev = device or (torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
    # gloo p2p on device memory is outside its supported envelope (a CUDA
    # send over gloo was observed to take a node down); transport through
    # CPU and copy back onto the requested device
    gloo_cuda = not _use_batched(pg) and (
        dev.type == "cuda"
        or (tensor_send_prev is not None and tensor_send_prev.is_cuda)
        or (tensor_send_next is not None and tensor_send_next.is_cuda)
    )
    wire_dev = torch.device("cpu") if gloo_cuda else dev
    send_prev = tensor_send_prev
    send_next = tensor_send_next
    if gloo_cuda:
        if send_prev is not None and send_prev.is_cuda:
            send_prev = send_prev.detach().cpu()
        if send_next is not None and send_next.is_cuda:
            send_next = send_next.detach().cpu()
    ops: List[dist.P2POp] = []
    if recv_prev:
        assert recv_shape is not None
        tensor_recv_prev = torch.empty(recv_shape, dtype=recv_dtype or torch.float32, device=wire_dev)
        ops.append(dist.P2POp(dist.irecv, tensor_recv_prev, peer=prev_rank, group=pg))
    if recv_next:
        assert recv_shape is not None
        tensor_recv_next = torch.empty(recv_shape, dtype=recv_dtype or torch.float32, device=wire_dev)
        ops.append(dist.P2POp(dist.irecv, tensor_recv_next, peer=next_rank, group=pg))
    if send_prev is not None:
        ops.append(dist.P2POp(dist.isend, send_prev.contiguous(), peer=prev_rank, group=pg))
    if send_next is not None:
        ops.append(dist.P2POp(dist.isend, send_next.contiguous(), peer=next_rank, group=pg))
    _run_p2p_ops(ops, pg)
    if gloo_cuda:
        if tensor_recv_prev is not None:
            tensor_recv_prev = tensor_recv_prev.to(dev)
        if tensor_recv_next is not None:
            tensor_recv_next = tensor_recv_next.to(dev)
    return tensor_recv_prev, tensor_recv_next


# ------------------------- public 8-call API -------------------------------
def _check_nan(t, what: str):
    """VESCALE_CHECK_NAN=1: raise on NaN/Inf in p2p payloads (reference
    p2p_communication.py:252 check_nan) — catches divergence at the stage
    boundary it crossed instead of steps later in the loss."""
    import os

    if t is None or not os.environ.get("VESCALE_CHECK_NAN"):
        return t
    bad = (~torch.isfinite(t.detach())).sum().item()
    if bad:
        rank = dist.get_rank() if dist.is_initialized() else 0
        raise FloatingPointError(
            f"[rank {rank}] {what}: {bad} non-finite elements in p2p tensor "
            f"shape={tuple(t.shape)} dtype={t.dtype}"
        )
    return t


def recv_forward(prev_rank, pg=None, shape=None, dtype=None, device=None):
    """Receive activations from the previous stage; counterpart of the
    shape handshake in send_forward."""
    t, _ = _communicate(recv_prev=True, prev_rank=prev_rank, pg=pg,
                        recv_shape=shape, recv_dtype=dtype, device=device)
    return _check_nan(t, "recv_forward")


def send_forward(t, next_rank, pg=None, handshake=True):
    if handshake:
        _communicate_shapes(t, None, next_rank, pg)
    _communicate(tensor_send_next=t, next_rank=next_rank, pg=pg)


def recv_backward(next_rank, pg=None, shape=None, dtype=None, device=None):
    _, g = _communicate(recv_next=True, next_rank=next_rank, pg=pg,
                        recv_shape=shape, recv_dtype=dtype, device=device)
    return _check_nan(g, "recv_backward")


def send_backward(g, prev_rank, pg=None, handshake=True):
    if handshake:
        _communicate_shapes(g, None, prev_rank, pg)
    _communicate(tensor_send_prev=g, prev_rank=prev_rank, pg=pg)


def send_forward_recv_backward(t, next_rank, pg=None, shape=None, dtype=None, device=None):
    if shape is None:
        meta = _communicate_shapes(t, next_rank, next_rank, pg)
        if meta:
            shape, dtype = meta
    _, g = _communicate(tensor_send_next=t, recv_next=True, next_rank=next_rank,
                        pg=pg, recv_shape=shape, recv_dtype=dtype, device=device)
    return g


def send_backward_recv_forward(g, prev_rank, pg=None, shape=None, dtype=None, device=None):
    if shape is None:
        meta = _communicate_shapes(g, prev_rank, prev_rank, pg)
        if meta:
            shape, dtype = meta
    t, _ = _communicate(tensor_send_prev=g, recv_prev=True, prev_rank=prev_rank,
                        pg=pg, recv_shape=shape, recv_dtype=dtype, device=device)
    return t
