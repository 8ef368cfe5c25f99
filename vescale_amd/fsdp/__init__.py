from .engine import FSDP, FSDPUnit
from .optimizer import FlatAdamW

__all__ = ["FSDP", "FSDPUnit", "FlatAdamW"]
