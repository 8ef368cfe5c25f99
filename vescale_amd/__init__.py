"""vescale_amd — a brand-new MI355X-native eager-mode SPMD distributed LLM
training framework with the capabilities of volcengine/veScale.

Layers (mirrors SURVEY.md §1, built MI355X-first):
  L1  dtensor/     DTensor runtime (placements, dispatch, redistribute -> RCCL)
  L3  fsdp/ dmodule/ ddp/ optim/ pipe/ moe/   parallelism engines
  L2  checkpoint/ ndtimeline/ emulator/ debug/  services
  L4  dmp/ plan    planning
  L0  ops/         hand-written CDNA4 HIP kernels (gfx950)
"""
from .dtensor import (  # noqa: F401
    DeviceMesh,
    DTensor,
    DTensorSpec,
    InterleavedShard,
    Partial,
    Placement,
    RaggedShard,
    Replicate,
    Shard,
    distribute_tensor,
    from_local,
    init_device_mesh,
    normalize_placements,
    redistribute_dtensor,
    to_local,
    vescale_all_gather,
    vescale_all_reduce,
    vescale_reduce_scatter,
)

# top-level API parity with the reference's `import vescale` surface
# (legacy/vescale/__init__.py): the whole framework is reachable from the
# package root.
from . import checkpoint  # noqa: F401,E402
from .ddp.distributed_data_parallel import DistributedDataParallel  # noqa: F401,E402
from .dmodule.api import (  # noqa: F401,E402
    PlacementsInterface,
    is_dmodule,
    parallelize_module,
)
from .dmp import (  # noqa: F401,E402
    auto_parallelize_module,
    get_plan_overriding_policy,
    set_plan_overriding_policy,
)
from .initialize.deferred_init import (  # noqa: F401,E402
    deferred_init,
    is_deferred,
    materialize_dparameter,
    materialize_dtensor,
)
from .optim.base_optimizer import BasicOptimizer, BasicOptimizerHook  # noqa: F401,E402
from .optim.distributed_optimizer import DistributedOptimizer  # noqa: F401,E402

__version__ = "0.1.0"
