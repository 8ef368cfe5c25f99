from .deferred_init import (
    deferred_init,
    is_deferred,
    materialize_dparameter,
    materialize_dmodule,
    materialize_dtensor,
    materialize_module,
)

__all__ = [
    "deferred_init",
    "is_deferred",
    "materialize_dtensor",
    "materialize_dparameter",
    "materialize_dmodule",
    "materialize_module",
]
