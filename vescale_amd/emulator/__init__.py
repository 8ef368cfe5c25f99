from .distributed import EmulatorProcessGroup, init_emulator
from .all_reduce import (
    run_direct_all_reduce,
    run_ring_all_reduce,
    run_tree_all_reduce,
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
    "double_binary_trees",
    "emu_all_gather",
    "emu_reduce_scatter",
    "emu_all_to_all",
]
