from .api import parallelize_module, PlacementsInterface

__all__ = ["parallelize_module", "PlacementsInterface"]
