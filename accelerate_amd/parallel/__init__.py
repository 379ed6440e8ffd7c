from .ddp import DistributedDataParallelEngine
from .ep import ExpertMLP, ExpertParallelMoE, balance_loss
from .pp import PipelineParallelEngine, split_into_stages

__all__ = [
    "DistributedDataParallelEngine",
    "ExpertMLP",
    "ExpertParallelMoE",
    "PipelineParallelEngine",
    "balance_loss",
    "split_into_stages",
]
