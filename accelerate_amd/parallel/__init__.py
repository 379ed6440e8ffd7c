from .ddp import DistributedDataParallelEngine
from .ep import ExpertMLP, ExpertParallelMoE, balance_loss

__all__ = [
    "DistributedDataParallelEngine",
    "ExpertMLP",
    "ExpertParallelMoE",
    "balance_loss",
]
