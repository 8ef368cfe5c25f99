"""RCCL algorithm / protocol selection + chunk-size accounting.

Parity: legacy/vescale/emulator/calculate_chunk_size.py:26-108
(`topo_get_algo_info`, `calcBytePerStep`, `compute_last_chunk_size`, NCCL
2.19.3 tuning tables) — re-derived for RCCL-over-xGMI rather than ported:
the constants are RCCL's defaults (NCCL_STEPS=8, 4 MiB per-channel simple
buffer, LL/LL128 step economics) and the size thresholds follow the same
shape as the NCCL tuner (latency-bound small messages prefer LL/tree,
bandwidth-bound large messages prefer Simple/ring; on xGMI the ring is
per-link bound so the ring cutover is LOWER than on NVSwitch).

This gives the emulator the same CHUNKED execution geometry as the real
collective: a ring all-reduce processes the buffer in loops of
`nranks * chunk` elements, and an element's position within its loop —
not within the whole buffer — decides which rank starts its reduction,
i.e. the exact floating-point addition order.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

# RCCL defaults (ncclTopoTuneModel / transport constants)
NCCL_STEPS = 8
BUFF_SIZE_SIMPLE = 1 << 22          # 4 MiB per channel
LL_MSG_OVERHEAD = 2                 # LL sends 4B data + 4B flag per 8B line
LL128_RATIO = 120 / 128             # LL128 carries 120B data per 128B line

PROTO_LL = "LL"
PROTO_LL128 = "LL128"
PROTO_SIMPLE = "Simple"

ALGO_RING = "ring"
ALGO_TREE = "tree"


def calc_byte_per_step(proto: str) -> int:
    """Payload bytes a channel moves per pipeline step (calcBytePerStep)."""
    step = BUFF_SIZE_SIMPLE // NCCL_STEPS  # 512 KiB
    if proto == PROTO_SIMPLE:
        return step
    if proto == PROTO_LL128:
        return int(step * LL128_RATIO)
    if proto == PROTO_LL:
        return step // (LL_MSG_OVERHEAD * 2)  # LL lines are 1/4 payload
    raise ValueError(proto)


def topo_get_algo_info(nbytes: int, nranks: int, nchannels: int = 1) -> Tuple[str, str, int]:
    """(algo, proto, nchannels) for an all-reduce of `nbytes` over `nranks`.

    Follows the NCCL tuner's structure: small = latency-dominated ->
    tree + LL; medium -> ring + LL128; large = bandwidth-dominated ->
    ring + Simple.  xGMI note: each of the 7 links is independent, so the
    ring (which keeps every link busy) wins earlier than on switched
    fabrics — the Simple-ring cutover is 512 KiB rather than a few MiB.
    """
    if nranks == 1:
        return ALGO_RING, PROTO_SIMPLE, 1
    if nbytes <= 64 << 10:
        return ALGO_TREE, PROTO_LL, max(1, nchannels)
    if nbytes <= 512 << 10:
        return ALGO_RING, PROTO_LL128, max(1, nchannels)
    return ALGO_RING, PROTO_SIMPLE, max(2, nchannels)


@dataclass
class RingChunkGeometry:
    """Per-loop segment layout of a chunked ring all-reduce."""
    numel: int
    nranks: int
    chunk_elems: int                 # elements per (rank, step) chunk
    loops: List[List[Tuple[int, int]]]  # loop -> per-chunk (offset, size)


def compute_last_chunk_size(numel: int, nranks: int, chunk_elems: int) -> int:
    """Elements in the final partial chunk (compute_last_chunk_size)."""
    loop_elems = nranks * chunk_elems
    tail = numel % loop_elems
    if tail == 0:
        return chunk_elems
    per = tail // nranks
    return per if per > 0 else tail


def ring_chunk_geometry(
    numel: int, elem_bytes: int, nranks: int, proto: str = PROTO_SIMPLE
) -> RingChunkGeometry:
    """Split a buffer the way the chunked ring processes it: loops of
    `nranks * chunk` elements; each loop is divided into `nranks` chunks
    (the last loop's chunks may shrink / be empty)."""
    chunk_elems = max(1, calc_byte_per_step(proto) // max(1, elem_bytes))
    loop_elems = nranks * chunk_elems
    loops: List[List[Tuple[int, int]]] = []
    off = 0
    while off < numel:
        this_loop = min(loop_elems, numel - off)
        base = this_loop // nranks
        rem = this_loop % nranks
        segs = []
        o = off
        for c in range(nranks):
            s = base + (1 if c < rem else 0)
            segs.append((o, s))
            o += s
        loops.append(segs)
        off += this_loop
    return RingChunkGeometry(numel, nranks, chunk_elems, loops)
