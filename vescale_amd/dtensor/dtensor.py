"""The DTensor wrapper subclass — single-device abstraction over a sharded /
replicated / partial global tensor.

Parity: legacy/vescale/dtensor/dtensor.py + api.py (from_local with
support_uneven, to_local, redistribute, full_tensor) and the new tree's
vescale/dtensor/_api.py.  Dispatch happens below autograd via
__torch_dispatch__ -> OpDispatcher (dispatch.py).
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ._dtensor_spec import DTensorSpec, make_spec
from .device_mesh import DeviceMesh
from .placement_types import (
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    TensorMeta,
)

__all__ = ["DTensor"]

aten = torch.ops.aten


def _stride_for(shape: Sequence[int]) -> Tuple[int, ...]:
    st, acc = [], 1
    for s in reversed(list(shape)):
        st.append(acc)
        acc *= s
    return tuple(reversed(st))


class DTensor(torch.Tensor):
    _local_tensor: torch.Tensor
    _spec: DTensorSpec

    __slots__ = ["_local_tensor", "_spec"]

    @staticmethod
    def __new__(cls, local_tensor: torch.Tensor, spec: DTensorSpec, *, requires_grad: bool):
        assert spec.tensor_meta is not None
        r = torch.Tensor._make_wrapper_subclass(
            cls,
            spec.tensor_meta.shape,
            strides=spec.tensor_meta.stride,
            dtype=local_tensor.dtype,
            device=local_tensor.device,
            layout=local_tensor.layout,
            requires_grad=requires_grad,
        )
        r._local_tensor = local_tensor
        r._spec = spec
        return r

    def __repr__(self):
        return f"DTensor(local={self._local_tensor.shape}, spec={self._spec})"

    def __tensor_flatten__(self):
        return ["_local_tensor"], (self._spec, self.requires_grad)

    @staticmethod
    def __tensor_unflatten__(inner, meta, outer_size, outer_stride):
        spec, requires_grad = meta
        return DTensor(inner["_local_tensor"], spec, requires_grad=requires_grad)

    @classmethod
    def __torch_dispatch__(cls, func, types, args=(), kwargs=None):
        from .dispatch import get_dispatcher

        return get_dispatcher().dispatch(func, args, kwargs or {})

    # eager-first torch.compile compat (reference patch #9's scope): dynamo
    # must never trace INTO DTensor construction — a wrapper-subclass
    # mid-__new__ breaks __tensor_flatten__-based fakeification.  Opaque
    # construction = graph break; the compiled region covers the local
    # compute between DTensor boundaries.  (Applied after class creation
    # below; no-op when torch.compiler is absent.)

    # ------------------------------------------------------------------
    @classmethod
    def _from_local_spec(
        cls, local: torch.Tensor, spec: DTensorSpec, requires_grad: bool = False
    ) -> "DTensor":
        return cls(local, spec, requires_grad=requires_grad)

    @staticmethod
    def from_local(
        local_tensor: torch.Tensor,
        device_mesh: DeviceMesh,
        placements: Sequence[Placement],
        *,
        run_check: bool = False,
        shape: Optional[torch.Size] = None,
        stride: Optional[Tuple[int, ...]] = None,
        support_uneven: bool = False,
    ) -> "DTensor":
        """Wrap a local tensor as a DTensor.  Global shape is inferred by
        multiplying sharded dims by mesh size unless `shape` is given; with
        support_uneven=True the true global extent of uneven Shard dims is
        agreed via an all_reduce of local sizes (reference
        legacy/vescale/dtensor/api.py:39, README.md:84)."""
        placements = tuple(placements)
        if shape is None:
            gshape = list(local_tensor.shape)
            for mesh_dim, p in enumerate(placements):
                w = device_mesh.size(mesh_dim)
                if isinstance(p, RaggedShard):
                    raise ValueError("from_local with RaggedShard requires explicit shape")
                if isinstance(p, (Shard, InterleavedShard)):
                    if support_uneven and not isinstance(p, InterleavedShard):
                        t = torch.tensor([local_tensor.shape[p.dim]], dtype=torch.int64)
                        dist.all_reduce(t, group=device_mesh.get_group(mesh_dim))
                        gshape[p.dim] = int(t.item())
                    else:
                        gshape[p.dim] = local_tensor.shape[p.dim] * w
            shape = torch.Size(gshape)
        if run_check:
            for mesh_dim, p in enumerate(placements):
                if isinstance(p, Replicate):
                    from . import _collective_utils as cc

                    t = local_tensor.contiguous()
                    cc.mesh_broadcast(t, device_mesh, mesh_dim)
                    local_tensor = t
        tm = TensorMeta(shape, stride or _stride_for(shape), local_tensor.dtype)
        spec = DTensorSpec(device_mesh, placements, tm)
        return _FromLocal.apply(local_tensor, spec)

    def to_local(self, *, grad_placements=None) -> torch.Tensor:
        """Local tensor of this rank.  grad_placements (reference
        _api.py:410 kwarg): declare the layout the GRADIENT flowing back
        into this boundary will have (e.g. Partial("sum") when downstream
        produces a partial-sum), overriding the default assumption that it
        matches this DTensor's placements."""
        if grad_placements is not None and not isinstance(grad_placements, tuple):
            grad_placements = tuple(grad_placements)
        return _ToLocal.apply(self, grad_placements)

    @property
    def device_mesh(self) -> DeviceMesh:
        return self._spec.mesh

    @property
    def placements(self) -> Tuple[Placement, ...]:
        return self._spec.placements

    def redistribute(
        self,
        device_mesh: Optional[DeviceMesh] = None,
        placements: Optional[Sequence[Placement]] = None,
        *,
        async_op: bool = False,
    ) -> "DTensor":
        from .redistribute import Redistribute

        assert device_mesh is None or device_mesh == self._spec.mesh, (
            "cross-mesh redistribute not supported here"
        )
        assert placements is not None
        return Redistribute.apply(self, tuple(placements), async_op)

    def full_tensor(self, *, grad_placements=None) -> torch.Tensor:
        rep = self.redistribute(placements=[Replicate()] * self._spec.mesh.ndim)
        return rep.to_local(grad_placements=grad_placements)

    # convenience parity helpers -----------------------------------------
    def local_shape(self) -> Tuple[int, ...]:
        return tuple(self._local_tensor.shape)

    def local_offsets(self) -> Tuple[int, ...]:
        return self._spec.local_offsets()


class _FromLocal(torch.autograd.Function):
    @staticmethod
    def forward(ctx, local: torch.Tensor, spec: DTensorSpec):
        ctx.spec = spec
        return DTensor(
            local.detach() if local.requires_grad else local,
            spec,
            requires_grad=local.requires_grad,
        )

    @staticmethod
    def backward(ctx, grad_output: "DTensor"):
        # grad wrt the local tensor: if grad came back Partial on a dim where
        # input was Replicate, reduce it (matches reference from_local grad)
        spec = ctx.spec
        g = grad_output
        if isinstance(g, DTensor):
            tgt = tuple(
                Replicate() if (isinstance(gp, Partial) and not isinstance(sp, Partial)) else sp
                for gp, sp in zip(g._spec.placements, spec.placements)
            )
            if tgt != g._spec.placements:
                g = g.redistribute(placements=tgt)
            return g._local_tensor, None
        return g, None


class _ToLocal(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dtensor: "DTensor", grad_placements=None):
        ctx.spec = dtensor._spec
        ctx.grad_placements = grad_placements
        lt = dtensor._local_tensor
        return lt.view_as(lt) if lt.requires_grad else lt

    @staticmethod
    def backward(ctx, grad_output: torch.Tensor):
        spec = ctx.spec
        if isinstance(grad_output, DTensor):
            # mixed plain/DTensor ops can hand a DTensor grad to the plain
            # side of a to_local boundary; reduce any Partial dims first
            # (skipping this re-wrapped an UNREDUCED partial and corrupted
            # every upstream weight grad — caught by the 4D PPxDPxTP test)
            g = grad_output
            tgt = tuple(
                Replicate() if isinstance(p, Partial) else p
                for p in g._spec.placements
            )
            if tgt != g._spec.placements:
                g = g.redistribute(placements=tgt)
            grad_output = g._local_tensor
        if ctx.grad_placements is not None:
            # caller-declared grad layout (reference to_local kwarg): the
            # plain grad coming back IS laid out per these placements
            grad_placements = tuple(ctx.grad_placements)
        else:
            grad_placements = tuple(
                Replicate() if isinstance(p, Partial) else p for p in spec.placements
            )
        tm = TensorMeta(spec.tensor_meta.shape, spec.tensor_meta.stride, grad_output.dtype)
        gspec = DTensorSpec(spec.mesh, grad_placements, tm)
        return (
            DTensor(grad_output, gspec, requires_grad=grad_output.requires_grad),
            None,
        )


_dynamo_disable = getattr(getattr(torch, "compiler", None), "disable", None)
if _dynamo_disable is not None:  # torch.compile present: opaque construction
    DTensor.__new__ = staticmethod(_dynamo_disable(DTensor.__new__))
