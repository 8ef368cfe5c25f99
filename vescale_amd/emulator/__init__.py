from .distributed import EmulatorProcessGroup, init_emulator
from .all_reduce import (
    run_direct_all_reduce,
    run_ring_all_reduce,
    run_tree_all_reduce,
)
from .calculate_chunk_size import (
    calc_byte_per_step,
    compute_last_chunk_size,
    ring_chunk_geometry,
    topo_get_algo_info,
)
from .graph_dump import (
    TopoGraph,
    parse_graph_dump,
    predict_time_us,
    select_algo_proto,
)
from .topo import double_binary_trees, ring_order
from .mesh_collectives import (
    emu_all_gather,
    emu_all_to_all,
    emu_reduce_scatter,
)

__all__ = [
    "init_emulator",
    "EmulatorProcessGroup",
    "run_ring_all_reduce",
    "run_tree_all_reduce",
    "run_direct_all_reduce",
    "ring_order",
    "topo_get_algo_info",
    "calc_byte_per_step",
    "compute_last_chunk_size",
    "ring_chunk_geometry",
    "double_binary_trees",
    "TopoGraph",
    "parse_graph_dump",
    "predict_time_us",
    "select_algo_proto",
    "emu_all_gather",
    "emu_reduce_scatter",
    "emu_all_to_all",
]
