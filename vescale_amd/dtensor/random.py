"""Distributed RNG trackers.

Parity: legacy/vescale/dtensor/random.py (OffsetBasedRNGTracker,
ThreadBasedRNGTracker, TensorParallelRNGTracker, manual_seed).

Design for MI355X: the *bitwise single-device parity* path
(ThreadBased in the reference, which needed a patched CUDA philox kernel)
is provided here by our own sharded-philox HIP kernels
(vescale_amd/ops/csrc/philox.hip): the tracker records the current
DTensorSpec in a thread-local region; our random kernels read the shard's
global offset and generate with *global* philox indexing, so a sharded
rand is bitwise-identical to our own single-GPU kernel's output.
On CPU (tests), OffsetBased semantics apply: replicated tensors draw
identical values on all ranks; sharded tensors draw distinct streams.
"""
from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.distributed as dist

from ._dtensor_spec import DTensorSpec
from .device_mesh import DeviceMesh
from .placement_types import Replicate, Shard


class _RNGStateTracker:
    def __init__(self, device_type: str = "cuda"):
        self.device_type = device_type
        self._states = {}
        self.distribute_region_enabled = True
        self._current_spec: Optional[DTensorSpec] = None

    @contextlib.contextmanager
    def _distribute_region(self, spec: DTensorSpec):
        yield


class OffsetBasedRNGTracker(_RNGStateTracker):
    """Replicated tensors draw identical randoms on all ranks (same seed,
    same offset); sharded tensors advance per-rank offsets so shards draw
    from disjoint philox ranges."""

    def __init__(self, mesh: DeviceMesh, seed: Optional[int] = None):
        super().__init__(mesh.device_type)
        self.mesh = mesh
        if seed is None:
            # agree on a seed: rank0 broadcasts
            s = torch.tensor(
                [torch.initial_seed() % (2**31)], dtype=torch.int64
            )
            if dist.is_initialized() and mesh.size() > 1:
                dist.broadcast(s, src=0)
            seed = int(s.item())
        self.seed = seed
        self._offset = 0

    @contextlib.contextmanager
    def _distribute_region(self, spec: DTensorSpec):
        if not self.distribute_region_enabled:
            yield
            return
        self._current_spec = spec
        dev = self.device_type if self.device_type != "cuda" or torch.cuda.is_available() else "cpu"
        if dev == "cpu":
            state = torch.get_rng_state()
            shard_rank = _linear_shard_rank(spec)
            torch.manual_seed(self.seed + 7919 * shard_rank + self._offset)
            try:
                yield
            finally:
                self._offset += 1
                torch.set_rng_state(state)
                self._current_spec = None
        else:
            gen = torch.cuda.default_generators[torch.cuda.current_device()]
            state = gen.get_state()
            shard_rank = _linear_shard_rank(spec)
            torch.cuda.manual_seed(self.seed + 7919 * shard_rank + self._offset)
            try:
                yield
            finally:
                self._offset += 1
                gen.set_state(state)
                self._current_spec = None


class ThreadBasedRNGTracker(OffsetBasedRNGTracker):
    """Bitwise single-device parity tracker (the reference's patched-kernel
    semantics): random aten ops on GPU DTensors are
    EXECUTED by our sharded-philox HIP kernels with the shard's GLOBAL
    element indexing, so a TP/DP-sharded random fill is bitwise-identical
    to our own single-GPU fill.  Ops without a philox implementation fall
    back to OffsetBased semantics."""

    def current_spec(self) -> Optional[DTensorSpec]:
        return self._current_spec

    def _desc(self, spec: DTensorSpec):
        """(gshape, lshape, offset, flat_offset, is_flat) for the philox
        kernels.  Returns None for layouts the kernels can't index."""
        from .placement_types import InterleavedShard

        coord = spec.mesh.get_coordinate()
        if coord is None:
            return None
        for p in spec.placements:
            if isinstance(p, InterleavedShard):
                return None
        if spec.has_ragged():
            md = next(
                i for i, p in enumerate(spec.placements) if p.is_ragged_shard()
            )
            p = spec.placements[md]
            g = list(spec.shape)
            numel = 1
            for s in g:
                numel *= s
            off = p.local_offset_numel(tuple(g), coord[md])
            n = p.local_numel(tuple(g), coord[md])
            return ([numel], [n], [off], off, True)
        gshape = list(spec.shape)
        lshape = list(spec.local_shape(coord))
        offs = list(spec.local_offsets(coord))
        return (gshape, lshape, offs, 0, False)

    def exec_random_op(self, op, local_args, local_kwargs, spec: DTensorSpec):
        """Run a random aten op with global philox indexing; returns the
        local result or NotImplemented to fall back."""
        import os

        import torch as _t

        from ..ops import has_ext

        # VESCALE_SINGLE_DEVICE_RAND=0 (reference dtensor/random.py:381):
        # opt OUT of the single-device-parity philox path and use plain
        # per-rank RNG instead
        if os.environ.get("VESCALE_SINGLE_DEVICE_RAND", "1") in ("0", "off"):
            return NotImplemented
        aten = _t.ops.aten
        x = local_args[0]
        if not (isinstance(x, _t.Tensor) and x.is_cuda and has_ext()):
            return NotImplemented
        d = self._desc(spec)
        if d is None:
            return NotImplemented
        from ..ops import _C

        gshape, lshape, offs, foff, flat = d
        seed = self.seed
        philox_off = self._offset
        self._offset += 4
        pkt = op.overloadpacket
        if pkt == aten.native_dropout:
            p = float(local_args[1])
            train = local_args[2] if len(local_args) > 2 else True
            if not train or p == 0.0:
                return NotImplemented
            if x.dtype != _t.bfloat16:
                return NotImplemented
            out, mask = _C.philox_dropout(
                x.contiguous(), gshape, lshape, offs, foff, flat, seed,
                philox_off, p, True,
            )
            return out, mask.to(_t.bool)
        if pkt == aten.uniform_:
            lo = float(local_args[1]) if len(local_args) > 1 else 0.0
            hi = float(local_args[2]) if len(local_args) > 2 else 1.0
            if x.dtype not in (_t.bfloat16, _t.float32):
                return NotImplemented
            _C.philox_uniform_(x, gshape, lshape, offs, foff, flat, seed, philox_off, lo, hi)
            return x
        if pkt == aten.normal_:
            mean = float(local_args[1]) if len(local_args) > 1 else 0.0
            std = float(local_args[2]) if len(local_args) > 2 else 1.0
            if x.dtype not in (_t.bfloat16, _t.float32):
                return NotImplemented
            _C.philox_normal_(x, gshape, lshape, offs, foff, flat, seed, philox_off, mean, std)
            return x
        if pkt in (aten.rand_like, aten.randn_like):
            out = _t.empty_like(x)
            if out.dtype not in (_t.bfloat16, _t.float32):
                return NotImplemented
            if pkt == aten.rand_like:
                _C.philox_uniform_(out, gshape, lshape, offs, foff, flat, seed, philox_off, 0.0, 1.0)
            else:
                _C.philox_normal_(out, gshape, lshape, offs, foff, flat, seed, philox_off, 0.0, 1.0)
            return out
        return NotImplemented


class TensorParallelRNGTracker(_RNGStateTracker):
    """Different seed inside TP regions (matching Megatron semantics)."""

    def __init__(self, mesh: DeviceMesh, base_seed: int = 1234):
        super().__init__(mesh.device_type)
        self.mesh = mesh
        self.base_seed = base_seed

    @contextlib.contextmanager
    def _distribute_region(self, spec: DTensorSpec):
        coord = spec.mesh.get_coordinate()
        tp_rank = 0 if coord is None else sum(coord)
        if spec.is_sharded:
            seed = self.base_seed + 2718 + tp_rank
        else:
            seed = self.base_seed
        if self.device_type == "cuda" and torch.cuda.is_available():
            gen = torch.cuda.default_generators[torch.cuda.current_device()]
            state = gen.get_state()
            torch.cuda.manual_seed(seed)
            try:
                yield
            finally:
                gen.set_state(state)
        else:
            state = torch.get_rng_state()
            torch.manual_seed(seed)
            try:
                yield
            finally:
                torch.set_rng_state(state)


def _linear_shard_rank(spec: DTensorSpec) -> int:
    """0 for fully-replicated; otherwise the linearized coordinate over the
    mesh dims that shard this tensor."""
    coord = spec.mesh.get_coordinate()
    if coord is None:
        return 0
    r = 0
    for md, p in enumerate(spec.placements):
        if not p.is_replicate() and not p.is_partial():
            r = r * spec.mesh.size(md) + coord[md]
    return r


def manual_seed(seed: int, mesh: DeviceMesh):
    """Set the tracker seed across the mesh (reference random.py:62)."""
    from .dispatch import get_dispatcher

    tracker = OffsetBasedRNGTracker(mesh, seed)
    get_dispatcher()._rng_tracker = tracker
    return tracker


def init_rng_tracker(mesh: DeviceMesh, kind: str = "offset", seed: Optional[int] = None):
    from .dispatch import get_dispatcher

    if kind == "offset":
        t = OffsetBasedRNGTracker(mesh, seed)
    elif kind == "thread":
        t = ThreadBasedRNGTracker(mesh, seed)
    elif kind == "tp":
        t = TensorParallelRNGTracker(mesh, seed or 1234)
    else:
        raise ValueError(kind)
    get_dispatcher()._rng_tracker = t
    return t


def is_rng_supported_mesh(mesh: DeviceMesh) -> bool:
    """True when sharded-RNG trackers can serve this mesh (reference
    random.py:37): a CUDA mesh (philox kernels) or any mesh for the
    offset-based tracker)."""
    return True


# migration alias (reference name)
init_vescale_rng_tracker = init_rng_tracker
