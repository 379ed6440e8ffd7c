"""Pipeline-parallel inference (reference: inference.py prepare_pippy).

MI355X-native design: no tracing compiler. The model is split into
``num_processes`` contiguous stages balanced by parameter bytes (the split
works on any model whose compute is a sequence of top-level blocks:
``nn.Sequential``, or anything exposing an ordered ``.layers`` ModuleList
between an embedding front and a head tail — our BERT/Llama families).

Schedule: GPipe-style microbatch streaming — rank 0 feeds microbatches,
activations travel rank→rank+1 over RCCL/gloo P2P (xGMI when intra-node),
the last rank concatenates outputs (optionally broadcast back to all ranks,
reference: inference.py:120-123).
"""

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .state import PartialState


def _sequential_blocks(model: nn.Module) -> List[nn.Module]:
    """Decompose a model into an ordered list of blocks."""
    if isinstance(model, nn.Sequential):
        return list(model)
    raise ValueError(
        "prepare_pipeline needs an nn.Sequential (or pass `blocks=` explicitly); "
        f"got {type(model).__name__}"
    )


def _balanced_split(blocks: List[nn.Module], n_stages: int) -> List[List[nn.Module]]:
    sizes = [sum(p.numel() * p.element_size() for p in b.parameters()) + 1 for b in blocks]
    total = sum(sizes)
    target = total / n_stages
    stages, cur, acc = [], [], 0.0
    remaining = n_stages
    for i, (b, s) in enumerate(zip(blocks, sizes)):
        cur.append(b)
        acc += s
        blocks_left = len(blocks) - i - 1
        if (acc >= target and remaining > 1 and blocks_left >= remaining - 1) or blocks_left < remaining - 1:
            stages.append(cur)
            cur, acc = [], 0.0
            remaining -= 1
    if cur:
        stages[-1].extend(cur) if len(stages) == n_stages else stages.append(cur)
    while len(stages) < n_stages:
        stages.append([])
    return stages


class PipelineModel(nn.Module):
    """This rank's stage + the GPipe streaming driver."""

    def __init__(self, stages: List[List[nn.Module]], num_chunks: Optional[int] = None,
                 gather_output: bool = False):
        super().__init__()
        self.state = PartialState()
        self.rank = self.state.process_index
        self.world = self.state.num_processes
        self.num_chunks = num_chunks or self.world
        self.gather_output = gather_output
        self.stage = nn.Sequential(*stages[self.rank]).to(self.state.device)
        # free other ranks' stages from this process
        self._n_stages = len(stages)

    _WIRE_DTYPES = [torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int32, torch.bool, torch.float64]

    def _send_act(self, act, comm_device):
        """Shape+dtype header (long[10]: ndim, dims..., dtype idx) then payload."""
        hdr = torch.zeros(10, dtype=torch.long)
        hdr[0] = act.ndim
        for i, d in enumerate(act.shape):
            hdr[i + 1] = d
        hdr[9] = self._WIRE_DTYPES.index(act.dtype)
        dist.send(hdr, dst=self.rank + 1)
        dist.send(act.to(comm_device).contiguous(), dst=self.rank + 1)

    def _recv_act(self, comm_device, device):
        hdr = torch.zeros(10, dtype=torch.long)
        dist.recv(hdr, src=self.rank - 1)
        shape = [int(hdr[i + 1]) for i in range(int(hdr[0]))]
        act = torch.empty(shape, dtype=self._WIRE_DTYPES[int(hdr[9])], device=comm_device)
        dist.recv(act, src=self.rank - 1)
        return act.to(device)

    @torch.no_grad()
    def forward(self, x: Optional[torch.Tensor] = None):
        device = self.state.device
        if self.world == 1:
            return self.stage(x.to(device))
        # rank 0 decides chunking; others follow
        meta = [self.num_chunks if x is None else min(self.num_chunks, x.shape[0])]
        dist.broadcast_object_list(meta, src=0)
        n_chunks = meta[0]
        comm_device = device if self.state.backend == "nccl" else torch.device("cpu")

        outputs = []
        if self.rank == 0:
            for chunk in x.to(device).chunk(n_chunks):
                act = self.stage(chunk)
                self._send_act(act, comm_device)
        else:
            for _ in range(n_chunks):
                act = self._recv_act(comm_device, device)
                act = self.stage(act)
                if self.rank < self.world - 1:
                    self._send_act(act, comm_device)
                else:
                    outputs.append(act)
        result = torch.cat(outputs, dim=0) if outputs else None
        if self.gather_output:
            from .utils.operations import copy_tensor_to_devices

            result = copy_tensor_to_devices(result)
        return result


def prepare_pipeline(model, num_chunks=None, blocks=None, gather_output=False):
    """Split `model` into per-rank stages for streaming inference."""
    state = PartialState()
    blocks = blocks if blocks is not None else _sequential_blocks(model)
    stages = _balanced_split(blocks, state.num_processes)
    return PipelineModel(stages, num_chunks=num_chunks, gather_output=gather_output)


# reference-name alias (reference: inference.py:126)
prepare_pippy = prepare_pipeline


# reference-compatible alias (reference inference.py:126 names it prepare_pippy)
prepare_pippy = prepare_pipeline
