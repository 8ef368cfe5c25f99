"""OpDispatcher — the eager per-op hot loop.

Flow (parity: legacy/vescale/dtensor/dispatch.py:235-392, redesigned):
  1. bypass table (ops answered without propagation)
  2. custom handlers (ops needing eager communication or special unwrap)
  3. sharding propagation (rule tables, lru-cached on a schema key)
  4. redistribute inputs to the rule's target placements  [=> RCCL]
  5. run the local aten op                                [=> HIP kernels]
  6. wrap outputs

CPU overhead is the eager-mode tax (reference warns at
vescale/dtensor/_dispatch.py:253-258): the cache key is built from flat
tuples without pytree, and cache hits skip all rule logic.
"""
from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional

import torch

from ._dtensor_spec import DTensorSpec
from ._op_schema import OpSchema, OutputSharding
from .placement_types import Replicate, Shard, TensorMeta
from .redistribute import redistribute_local_tensor


def _implicit_redistribute_guard(op, src_spec, tgt):
    """VESCALE_DISABLE_REDISTRIBUTE (reference dtensor/README.md:96-98):
    when set, any IMPLICIT redistribution inside op dispatch raises so
    hidden communication can be found and planned away.  Explicit
    DTensor.redistribute() calls are unaffected (they don't go through
    the dispatcher's input-target path).  NOTE a deliberate default
    difference: the reference ships with implicit redistribution DISABLED
    by default (_diff.py:24 reads the env with default "1"); this
    framework allows it by default (friendlier eager UX) and the flag
    opts INTO strictness."""
    import os

    if os.environ.get("VESCALE_DISABLE_REDISTRIBUTE"):
        raise RuntimeError(
            f"implicit redistribute disabled (VESCALE_DISABLE_REDISTRIBUTE): "
            f"op {op} wants {tuple(tgt)} but input is {src_spec.placements}"
        )

aten = torch.ops.aten

_STRICT = os.environ.get("VESCALE_AMD_STRICT", "0") == "1"


def _is_dtensor(x) -> bool:
    from .dtensor import DTensor

    return isinstance(x, DTensor)


class OpDispatcher:
    def __init__(self) -> None:
        self._rules: Dict[Any, Callable[[OpSchema], OutputSharding]] = {}
        self._handlers: Dict[Any, Callable] = {}
        self._bypass: Dict[Any, Callable] = {}
        self._cache: Dict[Any, OutputSharding] = {}
        self._fast_cache: Dict[Any, OutputSharding] = {}
        self._cache_cap = 16384  # bound for dynamic-shape workloads
        self._rng_tracker = None
        self._random_ops = set()
        self._pre_patches: list = []
        self._post_patches: list = []

    # -- registration ----------------------------------------------------
    def register_rule(self, op, fn):
        for o in self._expand(op):
            self._rules[o] = fn

    def register_handler(self, op, fn):
        for o in self._expand(op):
            self._handlers[o] = fn

    def register_bypass(self, op, fn):
        for o in self._expand(op):
            self._bypass[o] = fn

    def register_random(self, op):
        for o in self._expand(op):
            self._random_ops.add(o)

    # decoupled special-op patching (parity: legacy _dispatch_patch.py:37
    # DispatchPrePatch / post patch + _dispatch_bypass.py): pre patches may
    # rewrite (op, args, kwargs) before propagation; post patches may
    # rewrite the result.
    def register_pre_patch(self, fn):
        self._pre_patches.append(fn)

    def register_post_patch(self, fn):
        self._post_patches.append(fn)

    @staticmethod
    def _expand(op):
        if isinstance(op, torch._ops.OpOverloadPacket):
            return [getattr(op, n) for n in op.overloads()]
        return [op]

    # -- main entry ------------------------------------------------------
    def dispatch(self, op, args, kwargs):
        for pre in self._pre_patches:
            r = pre(op, args, kwargs)
            if r is not None:
                op, args, kwargs = r
        res = self._dispatch_inner(op, args, kwargs)
        for post in self._post_patches:
            r = post(op, args, kwargs, res)
            if r is not None:
                res = r
        return res

    def _dispatch_inner(self, op, args, kwargs):
        from .dtensor import DTensor

        h = self._bypass.get(op)
        if h is not None:
            return h(op, args, kwargs)
        h = self._handlers.get(op)
        if h is not None:
            return h(self, op, args, kwargs)

        # ---- fast path: flat positional args, hashable kwargs, warm cache.
        # Skips schema construction entirely (the eager hot-loop tax the
        # reference warns about at vescale/dtensor/_dispatch.py:253-258).
        fast = not kwargs or all(
            isinstance(v, (int, float, bool, str, type(None), torch.dtype))
            for v in kwargs.values()
        )
        if fast:
            key = [op]
            locals_ = []
            specs = []
            kp = key.append
            for a in args:
                if isinstance(a, DTensor):
                    sp = a._spec
                    locals_.append(a._local_tensor)
                    specs.append(sp)
                    kp((sp.placements, tuple(sp.tensor_meta.shape), sp.tensor_meta.dtype))
                elif isinstance(a, torch.Tensor):
                    fast = False
                    break
                elif isinstance(a, (list, tuple)):
                    fast = False
                    break
                else:
                    kp(a)
            if fast and specs:
                if kwargs:
                    for k in kwargs:
                        kp((k, kwargs[k]))
                try:
                    entry = self._fast_cache.get(tuple(key))
                except TypeError:
                    entry = None
                if entry is not None:
                    sharding = entry
                    mesh = specs[0].mesh
                    if sharding.input_targets is not None:
                        for i, tgt in enumerate(sharding.input_targets):
                            if tgt is None or tuple(tgt) == specs[i].placements:
                                continue
                            _implicit_redistribute_guard(op, specs[i], tgt)
                            tgt_spec = DTensorSpec(mesh, tuple(tgt), specs[i].tensor_meta)
                            locals_[i] = redistribute_local_tensor(
                                locals_[i], specs[i], tgt_spec
                            )
                            specs[i] = tgt_spec
                    it = iter(locals_)
                    local_args = tuple(
                        next(it) if isinstance(a, DTensor) else a for a in args
                    )
                    if op in self._random_ops and self._rng_tracker is not None:
                        res = NotImplemented
                        if hasattr(self._rng_tracker, "exec_random_op"):
                            res = self._rng_tracker.exec_random_op(
                                op, local_args, kwargs, specs[0]
                            )
                        if res is NotImplemented:
                            with self._rng_tracker._distribute_region(specs[0]):
                                res = op(*local_args, **kwargs)
                    else:
                        res = op(*local_args, **kwargs)
                    return self._wrap(res, sharding.output_spec, args)
        # ---- slow path (also populates the fast cache) ----

        # flatten: collect DTensor args in stable order
        specs: List[DTensorSpec] = []
        locals_: List[torch.Tensor] = []
        key_parts: List[Any] = [op]

        # pre-scan for the mesh so plain tensors can wrap regardless of order
        mesh0 = None

        def scan(x):
            nonlocal mesh0
            if mesh0 is None and isinstance(x, DTensor):
                mesh0 = x._spec.mesh
            elif isinstance(x, (list, tuple)):
                for y in x:
                    scan(y)

        for a in args:
            scan(a)
        for a in kwargs.values():
            scan(a)
        self._mesh0 = mesh0

        def conv(x):
            if isinstance(x, DTensor):
                specs.append(x._spec)
                locals_.append(x._local_tensor)
                key_parts.append(
                    (x._spec.placements, tuple(x._spec.shape), x.dtype)
                )
                return x._spec
            if isinstance(x, torch.Tensor):
                # plain tensor mixed in: wrap scalars / matching shapes as Replicate
                return self._wrap_plain(x, specs, locals_, key_parts)
            if isinstance(x, (list, tuple)):
                return type(x)(conv(y) for y in x)
            key_parts.append(x if isinstance(x, (int, float, bool, str, type(None), torch.dtype)) else str(x))
            return x

        args_schema = tuple(conv(a) for a in args)
        kwargs_schema = {k: conv(v) for k, v in kwargs.items()}

        if not specs:
            return op(*args, **kwargs)

        mesh = specs[0].mesh
        schema = OpSchema(op, args_schema, kwargs_schema)

        try:
            cache_key = tuple(key_parts) + tuple(sorted(kwargs.keys()))
            sharding = self._cache.get(cache_key)
        except TypeError:
            cache_key, sharding = None, None
        if sharding is None:
            sharding = self._propagate(schema)
            if cache_key is not None:
                if len(self._cache) >= self._cache_cap:
                    # dynamic shapes (e.g. variable-seq inference) would
                    # grow the schema cache without bound; refill is cheap
                    self._cache.clear()
                    self._fast_cache.clear()
                self._cache[cache_key] = sharding
        # mirror into the fast cache when the signature is flat
        try:
            if (not kwargs or all(
                isinstance(v, (int, float, bool, str, type(None), torch.dtype))
                for v in kwargs.values()
            )) and not any(isinstance(a, (list, tuple)) for a in args):
                fkey = [op]
                plain_ok = True
                for a in args:
                    if isinstance(a, DTensor):
                        sp0 = a._spec
                        fkey.append((sp0.placements, tuple(sp0.tensor_meta.shape), sp0.tensor_meta.dtype))
                    elif isinstance(a, torch.Tensor):
                        plain_ok = False
                        break
                    else:
                        fkey.append(a)
                if plain_ok:
                    for k in kwargs:
                        fkey.append((k, kwargs[k]))
                    self._fast_cache[tuple(fkey)] = sharding
        except TypeError:
            pass

        # redistribute inputs
        if sharding.input_targets is not None:
            for i, tgt in enumerate(sharding.input_targets):
                if tgt is None or tuple(tgt) == specs[i].placements:
                    continue
                _implicit_redistribute_guard(op, specs[i], tgt)
                tgt_spec = DTensorSpec(mesh, tuple(tgt), specs[i].tensor_meta)
                locals_[i] = redistribute_local_tensor(locals_[i], specs[i], tgt_spec)
                specs[i] = tgt_spec

        # rebuild local args
        it = iter(locals_)

        def fill(x):
            if isinstance(x, DTensorSpec):
                return next(it)
            if isinstance(x, (list, tuple)):
                return type(x)(fill(y) for y in x)
            return x

        local_args = tuple(fill(a) for a in args_schema)
        local_kwargs = {k: fill(v) for k, v in kwargs_schema.items()}

        # run the local op (random ops inside the RNG tracker region; the
        # ThreadBased tracker EXECUTES them via the sharded-philox HIP
        # kernels for bitwise single-device parity)
        if op in self._random_ops and self._rng_tracker is not None:
            res = NotImplemented
            if hasattr(self._rng_tracker, "exec_random_op"):
                res = self._rng_tracker.exec_random_op(
                    op, local_args, local_kwargs, specs[0]
                )
            if res is NotImplemented:
                with self._rng_tracker._distribute_region(specs[0]):
                    res = op(*local_args, **local_kwargs)
        else:
            res = op(*local_args, **local_kwargs)
        global _dbg
        if _dbg is None:
            from ..debug.debug_log import DebugLogger as _dbg_cls

            _dbg = _dbg_cls
        if _dbg.enabled():
            _dbg.log_op(op)

        return self._wrap(res, sharding.output_spec, args)

    # --------------------------------------------------------------
    def _wrap_plain(self, x, specs, locals_, key_parts):
        if x.ndim == 0 or x.numel() <= 1:
            key_parts.append(("scalar", x.dtype))
            return x  # scalar tensors stay plain (local op broadcasts)
        # full-shape plain tensor participating with DTensors: implicit
        # Replicate wrap (legal under the SPMD contract: all ranks hold the
        # same value).  Mesh is taken from the first real DTensor arg; if
        # none seen yet, defer by wrapping lazily (specs non-empty check
        # later guarantees at least one DTensor in the call).
        mesh = self._mesh0
        if mesh is None:
            key_parts.append(("plain", tuple(x.shape), x.dtype))
            return x
        tm = TensorMeta(x.shape, tuple(x.stride()), x.dtype)
        sp = DTensorSpec(mesh, tuple(Replicate() for _ in range(mesh.ndim)), tm)
        specs.append(sp)
        locals_.append(x)
        key_parts.append(("plainR", tuple(x.shape), x.dtype))
        return sp

    def _propagate(self, schema: OpSchema) -> OutputSharding:
        op = schema.op
        rule = self._rules.get(op)
        if rule is None:
            rule = self._rules.get(op.overloadpacket)
        if rule is not None:
            out = rule(schema)
            if out is not None:
                return out
        # fallback: replicate everything
        if _STRICT:
            raise NotImplementedError(
                f"no sharding rule for {op} with {schema.specs}; "
                f"set VESCALE_AMD_STRICT=0 to allow replicate fallback"
            )
        mesh = schema.mesh
        rep = tuple(Replicate() for _ in range(mesh.ndim))
        targets = [rep for _ in schema.specs]
        return OutputSharding(
            output_spec=_REPLICATE_OUT, input_targets=targets
        )

    def _wrap(self, res, out_spec, orig_args):
        from .dtensor import DTensor

        if res is None:
            return None
        if out_spec is _REPLICATE_OUT:
            # infer replicate spec(s) from result shapes
            mesh = None
            for a in orig_args:
                if _is_dtensor(a):
                    mesh = a._spec.mesh
                    break
                if isinstance(a, (list, tuple)):
                    for y in a:
                        if _is_dtensor(y):
                            mesh = y._spec.mesh
                            break
            def wrap_rep(t):
                if isinstance(t, torch.Tensor):
                    tm = TensorMeta(t.shape, tuple(t.stride()), t.dtype)
                    sp = DTensorSpec(mesh, tuple(Replicate() for _ in range(mesh.ndim)), tm)
                    return DTensor(t, sp, requires_grad=t.requires_grad)
                return t
            if isinstance(res, (list, tuple)):
                return type(res)(wrap_rep(t) for t in res)
            return wrap_rep(res)

        if isinstance(out_spec, DTensorSpec):
            assert isinstance(res, torch.Tensor), f"{res} for spec"
            out_spec = _fix_dtype(out_spec, res)
            return DTensor(res, out_spec, requires_grad=res.requires_grad)
        if isinstance(out_spec, (list, tuple)):
            assert isinstance(res, (list, tuple))
            out = []
            for t, sp in zip(res, out_spec):
                if sp is None or not isinstance(t, torch.Tensor):
                    out.append(t)
                else:
                    out.append(DTensor(t, _fix_dtype(sp, t), requires_grad=t.requires_grad))
            return type(res)(out)
        return res

    def clear_cache(self):
        self._cache.clear()
        self._fast_cache.clear()


def _fix_dtype(spec: DTensorSpec, res: torch.Tensor) -> DTensorSpec:
    tm = spec.tensor_meta
    if tm is not None and tm.dtype != res.dtype:
        return DTensorSpec(spec.mesh, spec.placements, TensorMeta(tm.shape, tm.stride, res.dtype))
    return spec


class _ReplicateOut:
    """Sentinel: wrap every tensor output as fully Replicate."""


_REPLICATE_OUT = _ReplicateOut()

_dbg = None
_dispatcher: Optional[OpDispatcher] = None


def get_dispatcher() -> OpDispatcher:
    global _dispatcher
    if _dispatcher is None:
        _dispatcher = OpDispatcher()
        from .ops import register_all

        register_all(_dispatcher)
    return _dispatcher
