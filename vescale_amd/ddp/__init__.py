from .distributed_data_parallel import DistributedDataParallel
from .grad_buffer import Bucket, GradBuffer

__all__ = ["DistributedDataParallel", "GradBuffer", "Bucket"]
