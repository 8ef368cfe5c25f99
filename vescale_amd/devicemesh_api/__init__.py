from .api import VeDeviceMesh, VESCALE_DEVICE_MESH

__all__ = ["VeDeviceMesh", "VESCALE_DEVICE_MESH"]
