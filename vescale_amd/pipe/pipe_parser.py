"""fx-graph-based pipeline parser.

Parity: legacy/vescale/pipe/tracer.py:81-709 (ModelTracer) +
pipe_parser.py:46-652 (PipeParser: torch.fx graph -> stage subgraphs,
split by MANUAL split points) — GRAPH_EAGER mode.  MANUAL_EAGER
(module-list) splitting lives in pipe_stage.py; this parser handles
models that are not a flat module list.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch
import torch.fx as fx
import torch.nn as nn
from torch.fx.passes.split_module import split_module


class ModelTracer(fx.Tracer):
    """Tracer that treats listed module classes as leaves (so stage
    boundaries land between whole submodules)."""

    def __init__(self, leaf_classes: Sequence[str] = ()):
        super().__init__()
        self.leaf_classes = set(leaf_classes)

    def is_leaf_module(self, m: nn.Module, qualname: str) -> bool:
        if type(m).__name__ in self.leaf_classes:
            return True
        return super().is_leaf_module(m, qualname)


def parse_model_graph(
    model: nn.Module,
    *,
    leaf_classes: Sequence[str] = ("TransformerBlock", "Block", "MixtralBlock"),
) -> fx.GraphModule:
    tracer = ModelTracer(leaf_classes)
    graph = tracer.trace(model)
    return fx.GraphModule(model, graph)


def split_pipeline_point(fqn: str, split_points: Sequence[str]) -> int:
    """Partition index of a module fqn given ordered split points: nodes at
    or after split_points[i] belong to partition i+1."""
    part = 0
    for i, sp in enumerate(split_points):
        if fqn == sp or fqn.startswith(sp + "."):
            return i + 1
    return part


def construct_pipeline_split_graph(
    model: nn.Module,
    split_points: Sequence[str],
    *,
    leaf_classes: Sequence[str] = ("TransformerBlock", "Block", "MixtralBlock"),
) -> List[fx.GraphModule]:
    """Trace and split into len(split_points)+1 stage GraphModules.
    split_points: module fqns that START each new stage (in order)."""
    gm = parse_model_graph(model, leaf_classes=leaf_classes)
    order: Dict[str, int] = {}
    cur = 0
    # walk nodes in topo order; bump partition when a split-point module
    # (or an op consuming only later-partition values) appears
    node_part: Dict[fx.Node, int] = {}
    for node in gm.graph.nodes:
        if node.op == "call_module":
            p = 0
            for i, sp in enumerate(split_points):
                if node.target == sp or str(node.target).startswith(sp + "."):
                    p = i + 1
                    break
                elif _module_after(str(node.target), sp, model):
                    p = i + 1
            cur = max(cur, p)
        node_part[node] = cur

    def mod_partition(node):
        return node_part.get(node, 0)

    split = split_module(gm, model, mod_partition)
    stages = []
    for name, sub in split.named_children():
        if name.startswith("submod_"):
            stages.append(sub)
    return stages


def _module_after(target: str, split_point: str, model: nn.Module) -> bool:
    """True if `target` comes after `split_point` in module registration
    order (the pipeline's sequential order)."""
    names = [n for n, _ in model.named_modules()]
    try:
        return names.index(target) >= names.index(split_point)
    except ValueError:
        return False


# ---------------------------------------------------------------------------
# HuggingFace-aware tracing (reference: legacy/vescale/pipe/tracer.py:93,626
# hf_symbolic_trace / HFTracer).  transformers >= 5 removed utils.fx, so this
# is a native implementation: a fixed-signature wrapper module absorbs the
# **kwargs-style HF forward (fx cannot complete **kwargs concrete args), HF
# block classes are traced as leaves, and the masking helpers — which branch
# on tensor data — are identity-patched into graph leaves for the duration
# of the trace (fx.wrap semantics applied to already-imported references).
# ---------------------------------------------------------------------------
_HF_LEAF_SUFFIXES = (
    "DecoderLayer",
    "RotaryEmbedding",
    "Block",
    "Attention",
    # norms as leaves keep their weights inside the owning stage (otherwise
    # split_module lifts them to parent get_attrs passed across stages)
    "RMSNorm",
    "LayerNorm",
)


class HFModelTracer(ModelTracer):
    """ModelTracer that also treats any module class named like an HF
    transformer block (``*DecoderLayer`` etc.) as a leaf, so stage
    boundaries land between whole blocks."""

    def is_leaf_module(self, m: nn.Module, qualname: str) -> bool:
        if type(m).__name__.endswith(_HF_LEAF_SUFFIXES):
            return True
        return super().is_leaf_module(m, qualname)


class _PatchedGraphLeaves:
    """Temporarily replace the given functions — at EVERY module attribute
    that holds them by identity — with wrappers that record a call_function
    node when invoked with fx Proxies (and behave normally otherwise)."""

    def __init__(self, fns):
        self.fns = [f for f in fns if f is not None]
        self._sites = []

    @staticmethod
    def _wrap(orig):
        import functools

        from torch.fx.proxy import Proxy

        def wrapped(*args, **kwargs):
            flat = list(args) + list(kwargs.values())
            prox = [a for a in flat if isinstance(a, Proxy)]
            if not prox:
                return orig(*args, **kwargs)
            tracer = prox[0].tracer

            def resolve(a):
                """Non-tensor attribute proxies (e.g. `self.config`) are
                bound by VALUE into the recorded call: GraphModule's
                get_attr copy of a config object loses private state
                (transformers' _attn_implementation came back None and the
                mask builder silently returned no mask)."""
                if isinstance(a, Proxy) and a.node.op == "get_attr":
                    obj = tracer.root
                    try:
                        for part in str(a.node.target).split("."):
                            obj = getattr(obj, part)
                    except AttributeError:
                        return a
                    if not isinstance(obj, torch.Tensor):
                        return obj
                return a

            args = tuple(resolve(a) for a in args)
            kwargs = {k: resolve(v) for k, v in kwargs.items()}
            const_kwargs = {
                k: v for k, v in kwargs.items() if not isinstance(v, Proxy)
                and not isinstance(v, torch.Tensor)
            }
            live_kwargs = {k: v for k, v in kwargs.items() if k not in const_kwargs}
            if const_kwargs:
                target = functools.partial(orig, **const_kwargs)
                functools.update_wrapper(target, orig)
            else:
                target = orig
            return tracer.create_proxy("call_function", target, args, live_kwargs)

        wrapped.__name__ = getattr(orig, "__name__", "wrapped")
        return wrapped

    def __enter__(self):
        import sys

        tid = {id(f): f for f in self.fns}
        wrappers = {i: self._wrap(f) for i, f in tid.items()}
        for name, mod in list(sys.modules.items()):
            if mod is None:
                continue
            try:
                d = vars(mod)
            except Exception:
                continue
            for attr, val in list(d.items()):
                if id(val) in tid:
                    self._sites.append((mod, attr, val))
                    setattr(mod, attr, wrappers[id(val)])
        return self

    def __exit__(self, *exc):
        for mod, attr, val in self._sites:
            setattr(mod, attr, val)
        self._sites.clear()
        return False


def _hf_graph_leaf_fns():
    """Data-dependent helper functions inside transformers that must become
    graph leaves (they branch on tensor values)."""
    try:
        import transformers.masking_utils as mu
    except Exception:
        return []
    return [
        getattr(mu, n, None)
        for n in (
            "create_causal_mask",
            "create_sliding_window_causal_mask",
            "create_chunked_causal_mask",
            "create_masks_for_generate",
        )
    ]


def hf_symbolic_trace(
    model: nn.Module,
    input_names: Sequence[str] = ("input_ids",),
    forward_kwargs: Optional[dict] = None,
) -> fx.GraphModule:
    """Trace a HuggingFace-style model (kwargs forward, **kwargs catch-all,
    ModelOutput returns) into an fx GraphModule whose positional inputs are
    `input_names`.  `forward_kwargs` are extra constants bound at the call
    (default: use_cache=False, return_dict=False — tuple outputs trace
    cleanly; index [0] of the result is the logits/hidden state)."""
    extra = dict(forward_kwargs or {"use_cache": False, "return_dict": False})

    args = ", ".join(input_names)
    calls = ", ".join(f"{n}={n}" for n in input_names)
    ns: dict = {}
    exec(  # noqa: S102 — builds the fixed-arity forward fx requires
        f"def forward(self, {args}):\n"
        f"    return self.inner({calls}, **self._extra)\n",
        ns,
    )

    wrap_cls = type(
        "_HFTraceRoot",
        (nn.Module,),
        {"forward": ns["forward"]},
    )
    root = wrap_cls()
    nn.Module.__init__(root)
    root.inner = model
    root._extra = extra

    tracer = HFModelTracer()
    with _PatchedGraphLeaves(_hf_graph_leaf_fns()):
        graph = tracer.trace(root)
    return fx.GraphModule(root, graph)


def split_graph_by_parameters(
    gm: fx.GraphModule, num_stages: int
) -> List[fx.GraphModule]:
    """PARAMETERS split criterion (reference pipe_parser.py:146): assign the
    traced graph's nodes to `num_stages` contiguous partitions so that each
    partition holds ~equal parameter bytes.  Non-module nodes ride with the
    current partition; get_attr parameters count toward it too."""

    def node_bytes(n: fx.Node) -> int:
        if n.op == "call_module":
            try:
                sub = gm.get_submodule(str(n.target))
            except AttributeError:
                return 0
            return sum(p.numel() * p.element_size() for p in sub.parameters())
        if n.op == "get_attr":
            try:
                t = gm.get_parameter(str(n.target))
                return t.numel() * t.element_size()
            except AttributeError:
                return 0
        return 0

    nodes = list(gm.graph.nodes)
    sizes = [node_bytes(n) for n in nodes]
    total = sum(sizes)
    target = total / max(num_stages, 1)
    part_of: Dict[fx.Node, int] = {}
    acc = 0.0
    part = 0
    started = False  # current partition has at least one parametrized node
    for n, sz in zip(nodes, sizes):
        if sz > 0 and started and part < num_stages - 1:
            boundary = target * (part + 1)
            # break at whichever side of the boundary is closer: taking the
            # node may overshoot less than stopping short (never-overshoot
            # greedy strands small leading modules in their own stage)
            if acc >= boundary or (acc + sz) - boundary > boundary - acc:
                part += 1
                started = False
        if sz > 0:
            started = True
        acc += sz
        part_of[n] = part

    split = split_module(gm, gm, lambda n: part_of.get(n, 0))
    return linearize_stages(split)


class _StageAdapter(nn.Module):
    """Tuple-in/tuple-out wrapper enforcing the pipeline wire protocol:
    stage k receives the full wire (everything later stages still need),
    consumes its own inputs by position, and re-emits its outputs plus the
    surviving pass-throughs.  This is what lets a DAG-shaped split (values
    skipping stages, e.g. the causal mask and rotary tables produced in
    stage 0 but consumed by every later stage) run on a strictly
    stage-to-stage p2p pipeline — the reference's parser does the same
    threading inside its graph splitter."""

    def __init__(self, stage: nn.Module, in_sel, out_sel):
        super().__init__()
        self.stage = stage
        # in_sel entries: ("w", i) = wire idx i, ("a", name) = own attribute
        self.in_sel = list(in_sel)
        # out_sel entries: ("o", i) = stage output i, ("w", i) = wire idx i
        self.out_sel = list(out_sel)

    def forward(self, *wire):
        args = [
            wire[i] if kind == "w" else getattr(self, i)
            for kind, i in self.in_sel
        ]
        outs = self.stage(*args)
        if not isinstance(outs, tuple):
            outs = (outs,)
        res = tuple(
            outs[i] if kind == "o" else wire[i] for kind, i in self.out_sel
        )
        return res[0] if len(res) == 1 else res


def linearize_stages(split: fx.GraphModule) -> List[nn.Module]:
    """Turn split_module's DAG of submod_* calls into a linear chain of
    tuple-in/tuple-out stages with pass-throughs threaded.  Parent-lifted
    get_attr values (parameters/buffers the splitter kept on the root) are
    bound onto the consuming stage adapter so every stage is
    self-contained."""
    import operator

    nodes = list(split.graph.nodes)
    sub_nodes = [
        n
        for n in nodes
        if n.op == "call_module" and str(n.target).startswith("submod_")
    ]
    order = {n: i for i, n in enumerate(sub_nodes)}
    n_stages = len(sub_nodes)

    def is_tuple_node(n):
        return n in order and any(
            u.op == "call_function" and u.target is operator.getitem
            for u in n.users
        )

    # producing stage of every wire-able value (placeholders -> -1)
    prod_stage: Dict[fx.Node, int] = {}
    for n in nodes:
        if n.op == "placeholder":
            prod_stage[n] = -1
        elif n in order:
            prod_stage[n] = order[n]
        elif n.op == "call_function" and n.target is operator.getitem:
            src = n.args[0]
            if src in prod_stage:
                prod_stage[n] = prod_stage[src]

    def flat_args(n):
        out = []

        def walk(a):
            if isinstance(a, fx.Node):
                out.append(a)
            elif isinstance(a, (list, tuple)):
                for x in a:
                    walk(x)

        for a in n.args:
            walk(a)
        return out

    consumers: Dict[fx.Node, List[int]] = {}
    for sn in sub_nodes:
        for a in flat_args(sn):
            consumers.setdefault(a, []).append(order[sn])
    out_node = next(n for n in nodes if n.op == "output")
    for a in flat_args(out_node):
        consumers.setdefault(a, []).append(n_stages)  # sentinel: final output

    def wire_before(k):
        """Values produced before stage k and still needed at >= k, in
        graph order (the inter-stage tensor protocol)."""
        w = []
        for n in nodes:
            if n not in prod_stage or is_tuple_node(n):
                continue
            if prod_stage[n] < k and any(c >= k for c in consumers.get(n, [])):
                w.append(n)
        return w

    def out_index(n):
        if n.op == "call_function" and n.target is operator.getitem:
            return n.args[1]
        return 0

    stages: List[nn.Module] = []
    for k, sn in enumerate(sub_nodes):
        win = wire_before(k)
        wout = wire_before(k + 1)
        idx_in = {n: i for i, n in enumerate(win)}
        stage_mod = split.get_submodule(str(sn.target))
        in_sel, bound = [], {}
        for a in flat_args(sn):
            if a in idx_in:
                in_sel.append(("w", idx_in[a]))
            elif a.op == "get_attr":
                # parent-held parameter/buffer: bind it onto this adapter so
                # the stage is self-contained (moves with .to()/optimizers)
                name = str(a.target).replace(".", "_")
                val = split
                for part in str(a.target).split("."):
                    val = getattr(val, part)
                bound[name] = val
                in_sel.append(("a", name))
            else:
                raise RuntimeError(f"stage {k}: unroutable input {a.op} {a}")
        out_pos = {
            n: out_index(n) for n in nodes if prod_stage.get(n) == k and not is_tuple_node(n)
        }
        out_sel = [
            ("o", out_pos[n]) if n in out_pos else ("w", idx_in[n]) for n in wout
        ]
        ad = _StageAdapter(stage_mod, in_sel, out_sel)
        for name, val in bound.items():
            setattr(ad, name, val)
        stages.append(ad)
    return stages


def parse_huggingface_model(
    model: nn.Module,
    num_stages: int,
    input_names: Sequence[str] = ("input_ids",),
    forward_kwargs: Optional[dict] = None,
) -> List[fx.GraphModule]:
    """Trace an HF-style model and split it into `num_stages` pipeline
    stages balanced by parameter bytes: the GRAPH_EAGER entry point for
    models that are not flat module lists."""
    gm = hf_symbolic_trace(model, input_names, forward_kwargs)
    return split_graph_by_parameters(gm, num_stages)
