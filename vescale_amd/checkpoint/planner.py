"""Save/load planners: plan LRU cache + load-balanced dedup.

Parity: legacy/vescale/checkpoint/planner/vescale/vescale_planner.py:42-214
and planner/common.py:65-170 — (a) PlanLRUCache: repeated saves of the
same state-dict structure skip local-plan recomputation; (b) balanced
dedup: replicated items are assigned ROUND-ROBIN across the ranks that
hold them instead of always to the lowest rank (DCP default), spreading
write bandwidth across the node (reference checkpoint/README.md:11).
"""
from __future__ import annotations

import hashlib
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

import torch.distributed.checkpoint as dcp
from torch.distributed.checkpoint.default_planner import (
    DefaultLoadPlanner,
    DefaultSavePlanner,
)
from torch.distributed.checkpoint.planner import SavePlan


class PlanLRUCache:
    def __init__(self, capacity: int = 8):
        self._cache: "OrderedDict[str, SavePlan]" = OrderedDict()
        self.capacity = capacity
        self.hits = 0
        self.misses = 0

    def get(self, key: str) -> Optional[SavePlan]:
        if key in self._cache:
            self._cache.move_to_end(key)
            self.hits += 1
            return self._cache[key]
        self.misses += 1
        return None

    def put(self, key: str, plan: SavePlan):
        self._cache[key] = plan
        self._cache.move_to_end(key)
        while len(self._cache) > self.capacity:
            self._cache.popitem(last=False)


_GLOBAL_PLAN_CACHE = PlanLRUCache()


def _state_dict_key(state_dict) -> str:
    h = hashlib.sha1()
    for k in sorted(state_dict.keys()):
        v = state_dict[k]
        h.update(k.encode())
        if hasattr(v, "shape"):
            h.update(str(tuple(v.shape)).encode())
            h.update(str(v.dtype).encode())
            if hasattr(v, "_spec"):
                h.update(repr(v._spec.placements).encode())
    return h.hexdigest()


class VeScaleSavePlanner(DefaultSavePlanner):
    def __init__(self, *args, use_plan_cache: bool = True, **kwargs):
        super().__init__(*args, **kwargs)
        self.use_plan_cache = use_plan_cache
        self.plan_cache = _GLOBAL_PLAN_CACHE

    def create_local_plan(self) -> SavePlan:
        key = None
        if self.use_plan_cache:
            try:
                key = _state_dict_key(self.state_dict)
                cached = self.plan_cache.get(key)
                if cached is not None:
                    self.plan = cached
                    return cached
            except Exception:
                key = None
        plan = super().create_local_plan()
        if key is not None:
            self.plan_cache.put(key, plan)
        return plan

    def create_global_plan(self, all_plans: List[SavePlan]):
        all_plans = _balanced_dedup(all_plans)
        return super().create_global_plan(all_plans)


def _balanced_dedup(all_plans: List[SavePlan]) -> List[SavePlan]:
    """Assign each replicated WriteItem to ONE rank, round-robin over the
    holders (write-bandwidth load balancing; reference
    planner/common.py:92 custom_dedup_tensors)."""
    holders: Dict[Tuple, List[int]] = {}
    for rank, plan in enumerate(all_plans):
        for item in plan.items:
            key = (item.index.fqn, tuple(item.index.offset or ()))
            holders.setdefault(key, []).append(rank)
    assign: Dict[Tuple, int] = {}
    counter = 0
    for key, ranks in sorted(holders.items()):
        if len(ranks) == 1:
            assign[key] = ranks[0]
        else:
            assign[key] = ranks[counter % len(ranks)]
            counter += 1
    new_plans = []
    for rank, plan in enumerate(all_plans):
        items = [
            it
            for it in plan.items
            if assign[(it.index.fqn, tuple(it.index.offset or ()))] == rank
        ]
        new_plans.append(dcp.planner.SavePlan(items=items, storage_data=plan.storage_data, planner_data=plan.planner_data))
    return new_plans


class VeScaleLoadPlanner(DefaultLoadPlanner):
    pass
