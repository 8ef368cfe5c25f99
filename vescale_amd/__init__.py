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
)

__version__ = "0.1.0"
