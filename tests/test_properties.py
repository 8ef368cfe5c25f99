"""Property-based tests (hypothesis) for invariant-heavy utilities."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=60, deadline=None)
@given(
    shape=st.lists(st.integers(1, 7), min_size=1, max_size=4),
    data=st.data(),
)
def test_break_ragged_box_partitions(shape, data):
    """Boxes exactly tile the flat range [a, b), each flat-contiguous,
    with at most 2*ndim-1 boxes (reference _break_ragged_box contract)."""
    from vescale_amd.checkpoint.ragged_boxes import (
        box_flat_start,
        box_numel,
        break_ragged_box,
    )

    n = 1
    for s in shape:
        n *= s
    a = data.draw(st.integers(0, n))
    b = data.draw(st.integers(a, n))
    boxes = break_ragged_box(shape, a, b)
    assert len(boxes) <= max(1, 2 * len(shape) - 1)
    covered = 0
    pos = a
    for box in boxes:
        fs = box_flat_start(shape, box)
        assert fs == pos  # flat order, gap-free
        pos += box_numel(box)
        covered += box_numel(box)
    assert covered == b - a


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(1, 4000),
    w=st.integers(1, 8),
    chunk=st.integers(8, 512),
)
def test_chunked_ring_allreduce_correct(n, w, chunk):
    """Chunked ring all-reduce equals the direct sum (fp32 tolerance) and
    all ranks agree, for arbitrary sizes/world/chunk."""
    from vescale_amd.emulator import run_ring_all_reduce

    g = torch.Generator().manual_seed(n * 31 + w)
    bufs = [torch.randn(n, generator=g) for _ in range(w)]
    ref = sum(b.double() for b in bufs)
    out = run_ring_all_reduce([b.clone() for b in bufs], chunk_bytes=chunk * 4)
    for o in out:
        assert torch.equal(o, out[0])
    assert torch.allclose(out[0].double(), ref, atol=1e-3, rtol=1e-4)


@settings(max_examples=60, deadline=None)
@given(
    e=st.integers(1, 64),
    w=st.integers(1, 16),
    data=st.data(),
)
def test_load_balanced_allocator_valid(e, w, data):
    """Every expert assigned to exactly one rank; per-rank cap ceil(E/W)
    respected; heavy experts spread across ranks."""
    from vescale_amd.moe.experts_allocator import LoadBalancedExpertsAllocator

    a = LoadBalancedExpertsAllocator(e, w)
    counts = data.draw(
        st.lists(st.floats(0, 1e6, allow_nan=False), min_size=e, max_size=e)
    )
    a.update(counts)
    cap = -(-e // w)
    per_rank = [0] * w
    for ex in range(e):
        r = a.owner_of(ex)
        assert 0 <= r < w
        per_rank[r] += 1
    assert all(c <= cap for c in per_rank)
    assert sum(per_rank) == e


@settings(max_examples=30, deadline=None)
@given(
    numel=st.integers(1, 100_000),
    nranks=st.integers(1, 8),
)
def test_ring_chunk_geometry_covers(numel, nranks):
    from vescale_amd.emulator import ring_chunk_geometry

    geo = ring_chunk_geometry(numel, 4, nranks)
    covered = 0
    pos = 0
    for loop in geo.loops:
        for off, sz in loop:
            assert off == pos
            pos += sz
            covered += sz
    assert covered == numel


@settings(max_examples=60, deadline=None)
@given(
    rows=st.integers(1, 12),
    cols=st.integers(1, 12),
    nseg=st.integers(1, 3),
    data=st.data(),
)
def test_global_boxes_roundtrip(rows, cols, nseg, data):
    """TP-segmented global boxes cover exactly the local flat range, and
    mapping each box back through the segment table recovers the local
    coordinates (checkpoint/flat_state.py)."""
    from vescale_amd.checkpoint.flat_state import _global_boxes

    nseg = min(nseg, rows)
    # partition local rows into nseg segments with disjoint global rows
    cuts = sorted(
        data.draw(
            st.lists(
                st.integers(1, rows - 1), min_size=nseg - 1, max_size=nseg - 1,
                unique=True,
            )
        )
    ) if nseg > 1 else []
    bounds = [0] + cuts + [rows]
    segs = []
    g = 0
    gdim = 0
    for i in range(nseg):
        l0, l1 = bounds[i], bounds[i + 1]
        gap = data.draw(st.integers(0, 4))
        g += gap
        segs.append((l0, l1 - l0, g))
        g += l1 - l0
        gdim = g
    p = torch.zeros(rows, cols)
    p._tp_shard = (0, gdim, segs)
    n = rows * cols
    a = data.draw(st.integers(0, n))
    b = data.draw(st.integers(a, n))
    boxes = _global_boxes((rows, cols), a, b, p)
    # coverage + local-flat ordering
    pos = a
    for (goff, sz), fs in boxes:
        assert fs == pos
        nel = 1
        for x in sz:
            nel *= x
        # map global rows back to local rows through the segment table
        for gr in range(goff[0], goff[0] + sz[0]):
            hits = [l0 + (gr - g0) for l0, ln, g0 in segs if g0 <= gr < g0 + ln]
            assert len(hits) == 1
        pos += nel
    assert pos == b


@settings(max_examples=80, deadline=None)
@given(nwg=st.integers(1, 4096))
def test_xcd_bijective_remap(nwg):
    """The m204 bijective XCD remap used by gemm8.hip (and the simpler
    %8==0 form in the attention kernels) must be a bijection on
    [0, nwg) for ANY grid size — the original m157 form was not
    (guide ERRATA #11)."""
    q, r = nwg >> 3, nwg & 7
    seen = set()
    for f in range(nwg):
        xcd, off = f & 7, f >> 3
        g = (xcd * (q + 1) if xcd < r else r * (q + 1) + (xcd - r) * q) + off
        assert 0 <= g < nwg
        seen.add(g)
    assert len(seen) == nwg


def _t_redistribute_planner_soak(rank, ws):
    """Property soak of the order-aware redistribute planner (round-2 lkey
    logic): any (cur -> tgt) pair over {R, S(0), S(1), Partial, SRS∘S}
    on a 2x2 mesh must terminate and produce value-correct full tensors."""
    import itertools

    import torch

    from vescale_amd.dtensor import (
        Partial,
        Replicate,
        Shard,
        _StridedRaggedShard,
        distribute_tensor,
        init_device_mesh,
    )

    mesh = init_device_mesh("cpu", (2, 2))
    w = torch.arange(64, dtype=torch.float32).reshape(8, 8)
    srs = _StridedRaggedShard(dims=(0,), local_units=(1, 1), split_factor=2)
    cands = [
        [Replicate(), Replicate()],
        [Replicate(), Shard(0)],
        [Shard(0), Replicate()],
        [Shard(0), Shard(1)],
        [Shard(1), Shard(0)],
        [srs, Shard(0)],
        [srs, Shard(1)],
    ]
    for cur, tgt in itertools.product(cands, cands):
        d = distribute_tensor(w, mesh, cur)
        r = d.redistribute(placements=tgt)
        assert torch.equal(r.full_tensor(), w), (cur, tgt)
        # and back again
        rr = r.redistribute(placements=cur)
        assert torch.equal(rr.full_tensor(), w), (tgt, cur)


def test_redistribute_planner_soak():
    from tests.common import spawn

    spawn(4, _t_redistribute_planner_soak)
