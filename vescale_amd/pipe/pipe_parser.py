"""fx-graph-based pipeline parser.

Parity: legacy/vescale/pipe/tracer.py:81-709 (ModelTracer) +
pipe_parser.py:46-652 (PipeParser: torch.fx graph -> stage subgraphs,
split by MANUAL split points) — GRAPH_EAGER mode.  MANUAL_EAGER
(module-list) splitting lives in pipe_stage.py; this parser handles
models that are not a flat module list.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Type

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
