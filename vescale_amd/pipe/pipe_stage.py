"""Pipeline stage construction.

Parity: legacy/vescale/pipe/pipe_stage.py:64-563 (PipeModule,
construct_pipeline_stage, shared-embedding sync groups) + the MANUAL /
UNIFORM / PARAMETERS split criteria of pipe_parser.py — implemented over
module lists (MANUAL_EAGER mode); the fx-graph parser is in
pipe_parser.py.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist
import torch.nn as nn

from ..plan import PipelineParallelPlan, PipelineScheduleType, PipelineSplitMethodType


class PipeModule(nn.Module):
    """One rank's stage: a list of virtual chunks (>=1 modules) plus
    shared-parameter sync groups (tied embeddings across stages)."""

    def __init__(
        self,
        chunks: List[nn.Module],
        stage_id: int,
        n_stages: int,
        shared_param_groups: Optional[List[Dict]] = None,
    ):
        super().__init__()
        self.chunks = nn.ModuleList(chunks)
        self.stage_id = stage_id
        self.n_stages = n_stages
        self.shared_param_groups = shared_param_groups or []

    def forward(self, x, chunk: int = 0):
        return self.chunks[chunk](x)

    def parameters_of_chunk(self, chunk: int):
        return list(self.chunks[chunk].parameters())

    def sync_shared_params(self, pg_of_group: Optional[List] = None):
        """Allreduce gradients of tied parameters across their stage group
        (reference pipe_stage.py:235-246 + engine/pipe.py:229)."""
        for i, grp in enumerate(self.shared_param_groups):
            pg = grp.get("pg")
            p = grp.get("param")
            if pg is None or p is None:
                continue
            g = p.grad
            if g is None:
                g = torch.zeros_like(
                    p.data._local_tensor if hasattr(p.data, "_local_tensor") else p.data
                )
            local = g._local_tensor if hasattr(g, "_local_tensor") else g
            dist.all_reduce(local, group=pg)
            p.grad = g


def build_shared_module_group(
    stage: PipeModule,
    shared_fqns: Sequence[Sequence[str]],
    stage_of_fqn: dict,
    my_stage: int,
    stage_to_rank=None,
):
    """Build allreduce groups for TIED parameters living on different
    stages (reference pipe_stage.py:311 build_shared_module_group — e.g.
    input/output embeddings).  shared_fqns: groups of parameter fqns that
    are one logical weight; stage_of_fqn maps fqn -> owning stage.
    Creates one ProcessGroup per tied group (ALL ranks iterate every group
    — the collective new_group contract) and records the local param."""
    import torch.distributed as dist

    stage_to_rank = stage_to_rank or (lambda s: s)
    groups = []
    for fqns in shared_fqns:
        stages = sorted({stage_of_fqn[f] for f in fqns})
        ranks = [stage_to_rank(s) for s in stages]
        pg = dist.new_group(ranks=ranks) if dist.is_initialized() and len(ranks) > 1 else None
        entry = {"pg": pg, "param": None, "fqns": list(fqns)}
        if my_stage in stages:
            params = dict(stage.named_parameters())
            for f in fqns:
                if stage_of_fqn[f] == my_stage:
                    # fqn within the stage module namespace
                    short = f.split(".")[-1]
                    for n, p in params.items():
                        if n.endswith(f) or n.endswith(short):
                            entry["param"] = p
                            break
        groups.append(entry)
    stage.shared_param_groups = groups
    return groups


def uniform_split(modules: Sequence[nn.Module], n_parts: int) -> List[List[nn.Module]]:
    """Split a module list into n contiguous parts balanced by parameter
    count (reference PipelineSplitMethodType.PARAMETERS/UNIFORM)."""
    sizes = [sum(p.numel() for p in m.parameters()) for m in modules]
    total = sum(sizes)
    target = total / max(n_parts, 1)
    parts: List[List[nn.Module]] = [[] for _ in range(n_parts)]
    acc = 0.0
    idx = 0
    for m, s in zip(modules, sizes):
        if acc >= target * (idx + 1) and idx < n_parts - 1 and parts[idx]:
            idx += 1
        parts[idx].append(m)
        acc += s
    return parts


def construct_pipeline_stage(
    module_list: Sequence[nn.Module],
    plan: PipelineParallelPlan,
    stage_id: int,
    *,
    split_points: Optional[List[int]] = None,
) -> PipeModule:
    """module_list: the model's sequential units (e.g. [embed, block0, ...,
    blockN, head]).  Builds this rank's PipeModule of `virtual_chunks`
    chunks out of `num_stages * virtual_chunks` parts."""
    n_parts = plan.num_stages * plan.virtual_chunks
    if plan.split_method == PipelineSplitMethodType.MANUAL and split_points:
        assert len(split_points) == n_parts - 1
        bounds = [0] + list(split_points) + [len(module_list)]
        parts = [list(module_list[bounds[i] : bounds[i + 1]]) for i in range(n_parts)]
    else:
        parts = uniform_split(module_list, n_parts)
    zbv = (
        plan.schedule_type == PipelineScheduleType.ZERO_BUBBLE
        and plan.virtual_chunks == 2
    )
    chunks = []
    for ck in range(plan.virtual_chunks):
        if zbv:
            # V placement (zbv.py): chunk 0 descends the ranks, chunk 1
            # ascends — rank s hosts parts s and 2P-1-s
            idx = stage_id if ck == 0 else n_parts - 1 - stage_id
        else:
            idx = ck * plan.num_stages + stage_id
        part = parts[idx]
        chunks.append(nn.Sequential(*part) if len(part) != 1 else part[0])
    return PipeModule(chunks, stage_id, plan.num_stages)


def construct_stage_modules(
    module_list: Sequence[nn.Module],
    plan: PipelineParallelPlan,
    stage_id: int,
    *,
    split_points: Optional[List[int]] = None,
) -> List[nn.Module]:
    """The raw chunk modules this stage hosts (reference pipe_stage.py:249
    construct_stage_modules) — construct_pipeline_stage without the
    PipeModule wrapper."""
    return list(
        construct_pipeline_stage(
            module_list, plan, stage_id, split_points=split_points
        ).chunks
    )
