"""MI355X sharded-parameter engine (FSDP2-equivalent) — see SURVEY.md §2.9 N4.

Design (build plan §7 step 9): per-parameter shards resident in HBM3E,
forward all-gather / backward reduce-scatter over RCCL on xGMI with
prefetch streams, bf16 compute + fp32 master policy.

This module currently provides the scaffolding used by the rest of the
framework; the full engine lands later in the round.
"""

from typing import Iterable

import torch
import torch.nn as nn


class ShardedModel(nn.Module):
    """Placeholder wrapper type so isinstance checks across the framework are
    stable while the engine is being built."""

    def __init__(self, module: nn.Module):
        super().__init__()
        self.module = module

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)


def gather_full_state_dict(model: "ShardedModel"):
    return model.module.state_dict()


def fsdp_prepare(accelerator, args, device_placement):
    raise NotImplementedError(
        "The MI355X sharded-parameter engine (FSDP equivalent) is under construction this round; "
        "use the DDP path (default) for now."
    )
