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

from typing import List, Optional, Sequence, Tuple

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


def _meta_encode(t: torch.Tensor) -> torch.Tensor:
    m = torch.zeros(_MAX_DIMS + 2, dtype=torch.int64)
    m[0] = t.ndim
    for i, s in enumerate(t.shape):
        m[1 + i] = s
    m[_MAX_DIMS + 1] = _DTYPES.index(t.dtype)
    return m


def _meta_decode(m: torch.Tensor) -> Tuple[Tuple[int, ...], torch.dtype]:
    nd = int(m[0])
    shape = tuple(int(m[1 + i]) for i in range(nd))
    return shape, _DTYPES[int(m[_MAX_DIMS + 1])]


def _communicate_shapes(send_t: Optional[torch.Tensor], recv_from: Optional[int],
                        send_to: Optional[int], pg) -> Optional[Tuple]:
    """Exchange (shape, dtype) metadata before variable-shape transfers
    (reference :125)."""
    ops = []
    recv_buf = None
    if recv_from is not None:
        recv_buf = torch.zeros(_MAX_DIMS + 2, dtype=torch.int64)
        ops.append(dist.P2POp(dist.irecv, recv_buf, peer=recv_from, group=pg))
    if send_to is not None and send_t is not None:
        ops.append(dist.P2POp(dist.isend, _meta_encode(send_t), peer=send_to, group=pg))
    if ops:
        _run_p2p_ops(ops, pg)
    if recv_buf is not None:
        return _meta_decode(recv_buf)
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
    ops: List[dist.P2POp] = []
    if recv_prev:
        assert recv_shape is not None
        tensor_recv_prev = torch.empty(recv_shape, dtype=recv_dtype or torch.float32, device=dev)
        ops.append(dist.P2POp(dist.irecv, tensor_recv_prev, peer=prev_rank, group=pg))
    if recv_next:
        assert recv_shape is not None
        tensor_recv_next = torch.empty(recv_shape, dtype=recv_dtype or torch.float32, device=dev)
        ops.append(dist.P2POp(dist.irecv, tensor_recv_next, peer=next_rank, group=pg))
    if tensor_send_prev is not None:
        ops.append(dist.P2POp(dist.isend, tensor_send_prev.contiguous(), peer=prev_rank, group=pg))
    if tensor_send_next is not None:
        ops.append(dist.P2POp(dist.isend, tensor_send_next.contiguous(), peer=next_rank, group=pg))
    _run_p2p_ops(ops, pg)
    return tensor_recv_prev, tensor_recv_next


# ------------------------- public 8-call API -------------------------------
def recv_forward(prev_rank, pg=None, shape=None, dtype=None, device=None):
    """Receive activations from the previous stage; counterpart of the
    shape handshake in send_forward."""
    t, _ = _communicate(recv_prev=True, prev_rank=prev_rank, pg=pg,
                        recv_shape=shape, recv_dtype=dtype, device=device)
    return t


def send_forward(t, next_rank, pg=None, handshake=True):
    if handshake:
        _communicate_shapes(t, None, next_rank, pg)
    _communicate(tensor_send_next=t, next_rank=next_rank, pg=pg)


def recv_backward(next_rank, pg=None, shape=None, dtype=None, device=None):
    _, g = _communicate(recv_next=True, next_rank=next_rank, pg=pg,
                        recv_shape=shape, recv_dtype=dtype, device=device)
    return g


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
