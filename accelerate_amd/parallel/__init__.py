from .ddp import DistributedDataParallelEngine

__all__ = ["DistributedDataParallelEngine"]
