"""RCCL/NCCL topology-graph ingestion + latency/bandwidth tuning model.

Reference capability: legacy/vescale/emulator's `dump_nccl_graph_for_pg`
workflow (emulator/README.md:76-80 — run once with NCCL_GRAPH_DUMP_FILE
set, feed the dumped XML into the emulator so algo/proto/chunk selection
uses the REAL topology's nchannels/bandwidth/latency) and its
nccl/graph/tuning.py tables.  Re-designed for RCCL-over-xGMI:

- `parse_graph_dump(path_or_xml)` reads the XML that RCCL (same code path
  as NCCL's ncclTopoDumpGraphs) writes for NCCL_GRAPH_DUMP_FILE: a
  <graphs> root with <graph pattern=... nchannels=... speedintra=...
  speedinter=... latencyinter=...> elements whose <channel> children list
  the gpu order — i.e. the ACTUAL ring order per channel.
- `TopoGraph.ring(channel)` exposes that order, replacing the emulator's
  default identity ring so the emulated reduction order matches the
  machine's.
- `select_algo_proto(nbytes, nranks, topo)` is the tuner structure NCCL
  uses (predicted time = base latency + bytes / bw for each (algo, proto)
  pair; argmin), with RCCL-flavored constants for a fully-connected
  7-link xGMI node instead of NVSwitch.

No GPU is needed to exercise any of this; a dump captured on the target
node makes the model exact, and absent a dump the defaults match
calculate_chunk_size.topo_get_algo_info's thresholds.
"""
from __future__ import annotations

import os
import xml.etree.ElementTree as ET
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .calculate_chunk_size import (
    ALGO_RING,
    ALGO_TREE,
    PROTO_LL,
    PROTO_LL128,
    PROTO_SIMPLE,
)


@dataclass
class GraphSpec:
    """One <graph> element of the dump."""

    id: int
    pattern: str               # "ring" / "tree" / ... (nccl pattern enum name)
    nchannels: int
    speed_intra: float         # GB/s per channel inside the node
    speed_inter: float         # GB/s per channel across nodes
    latency_inter: float       # us
    type_intra: str = ""       # link type, e.g. "XGMI"/"NVL"/"PIX"
    channels: List[List[int]] = field(default_factory=list)  # gpu order per channel


@dataclass
class TopoGraph:
    graphs: Dict[str, GraphSpec] = field(default_factory=dict)

    def graph(self, pattern: str) -> Optional[GraphSpec]:
        return self.graphs.get(pattern)

    def ring(self, channel: int = 0) -> Optional[List[int]]:
        g = self.graph("ring")
        if g is None or not g.channels:
            return None
        return g.channels[channel % len(g.channels)]

    @property
    def nchannels(self) -> int:
        g = self.graph("ring") or next(iter(self.graphs.values()), None)
        return g.nchannels if g else 1

    def bw_intra(self, pattern: str = "ring") -> float:
        g = self.graph(pattern)
        return g.speed_intra * g.nchannels if g else 0.0


_PATTERN_NAMES = {
    # ncclTopoDumpGraphs writes the numeric pattern; RCCL builds name it too
    "0": "balanced_tree",
    "1": "split_tree",
    "2": "tree",
    "3": "ring",
    "4": "collnet_direct",
    "5": "collnet_chain",
    "6": "nvls",
}


def parse_graph_dump(path_or_xml: str) -> TopoGraph:
    """Parse an NCCL_GRAPH_DUMP_FILE XML (path or literal xml string)."""
    if os.path.exists(path_or_xml):
        root = ET.parse(path_or_xml).getroot()
    else:
        root = ET.fromstring(path_or_xml)
    topo = TopoGraph()
    for g in root.iter("graph"):
        pattern = g.get("pattern", "")
        pattern = _PATTERN_NAMES.get(pattern, pattern) or f"graph{g.get('id', 0)}"
        spec = GraphSpec(
            id=int(g.get("id", 0)),
            pattern=pattern,
            nchannels=int(g.get("nchannels", 1)),
            speed_intra=float(g.get("speedintra", 0.0)),
            speed_inter=float(g.get("speedinter", 0.0)),
            latency_inter=float(g.get("latencyinter", 0.0)),
            type_intra=g.get("typeintra", ""),
        )
        for ch in g.iter("channel"):
            order = [int(n.get("dev", n.get("rank", 0))) for n in ch]
            if order:
                spec.channels.append(order)
        topo.graphs[spec.pattern] = spec
    return topo


# ---------------------------------------------------------------------------
# tuning model (reference nccl/graph/tuning.py table structure, RCCL-flavored
# constants for one fully-connected xGMI node)
# ---------------------------------------------------------------------------
# base launch latency (us) per (algo, proto)
_BASE_LAT_US: Dict[Tuple[str, str], float] = {
    (ALGO_TREE, PROTO_LL): 4.4,
    (ALGO_TREE, PROTO_LL128): 5.6,
    (ALGO_TREE, PROTO_SIMPLE): 22.0,
    (ALGO_RING, PROTO_LL): 3.6,
    (ALGO_RING, PROTO_LL128): 5.6,
    (ALGO_RING, PROTO_SIMPLE): 16.0,
}
# per-hop link latency (us) on an intra-node xGMI fabric
_HOP_LAT_US: Dict[str, float] = {PROTO_LL: 0.5, PROTO_LL128: 0.8, PROTO_SIMPLE: 2.2}
# protocol payload efficiency (fraction of raw link bandwidth)
_PROTO_EFF: Dict[str, float] = {PROTO_LL: 0.25, PROTO_LL128: 0.85, PROTO_SIMPLE: 0.92}
# default per-link xGMI bandwidth (GB/s) when no dump is given
_XGMI_LINK_GBS = 153.0


def predict_time_us(
    nbytes: int,
    nranks: int,
    algo: str,
    proto: str,
    topo: Optional[TopoGraph] = None,
) -> float:
    """NCCL-tuner-shaped cost: base latency + hop latency * depth +
    transfer time over the algo's effective bandwidth."""
    if nranks <= 1:
        return 0.0
    g = topo.graph("ring") if topo else None
    nch = g.nchannels if g else 1
    link_gbs = g.speed_intra if g and g.speed_intra > 0 else _XGMI_LINK_GBS
    eff_bw = link_gbs * nch * _PROTO_EFF[proto]  # GB/s
    if algo == ALGO_RING:
        # 2(n-1)/n of the data crosses each link
        vol = 2.0 * (nranks - 1) / nranks * nbytes
        depth = nranks - 1
    else:  # tree: up + down, log2 depth, ~2x data volume
        import math

        vol = 2.0 * nbytes
        depth = max(1, int(math.ceil(math.log2(nranks))))
    lat = _BASE_LAT_US[(algo, proto)] + _HOP_LAT_US[proto] * depth
    return lat + (vol / (eff_bw * 1e3))  # bytes / (GB/s) -> ns/1e3 = us


def select_algo_proto(
    nbytes: int, nranks: int, topo: Optional[TopoGraph] = None
) -> Tuple[str, str, int]:
    """Pick the (algo, proto) minimizing predicted time — the tuner's
    actual structure, vs the fixed thresholds of topo_get_algo_info.
    Returns (algo, proto, nchannels)."""
    best = None
    # RCCL disables LL128 on CDNA: the protocol relies on 128-byte
    # single-shot write atomicity that xGMI does not guarantee.  A dump
    # that reports an XGMI intra-node link (or no dump at all, since this
    # emulator targets MI355X) therefore never selects LL128.
    g = topo.graph("ring") if topo else None
    xgmi = g is None or g.type_intra.upper() in ("", "XGMI")
    protos = (PROTO_LL, PROTO_SIMPLE) if xgmi else (PROTO_LL, PROTO_LL128, PROTO_SIMPLE)
    for algo in (ALGO_RING, ALGO_TREE):
        for proto in protos:
            t = predict_time_us(nbytes, nranks, algo, proto, topo)
            if best is None or t < best[0]:
                best = (t, algo, proto)
    nch = topo.nchannels if topo else 1
    return best[1], best[2], nch
