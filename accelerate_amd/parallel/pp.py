"""Pipeline-parallel TRAINING (GPipe schedule over P2P).

The reference implements pipeline parallelism for inference only and
explicitly raises for training (reference accelerator.py:795-799
NotImplementedError; inference lives in inference.py). This engine goes
beyond it: microbatched forward/backward with activation hand-off over
``dist.send``/``dist.recv`` — xGMI point-to-point on RCCL, plain TCP on
gloo (which is how the 2-process CPU oracle runs it).

Schedule (GPipe, all-forward-then-all-backward):

  stage r, microbatch m:   recv a[m] from r-1 → run stage → send to r+1
  ...all M microbatches...
  then in reverse:         recv g[m] from r+1 → backward → send input-grad

Gradients accumulate across microbatches in the stage's parameters; the
caller steps its optimizer once per ``train_step``. The LAST stage owns
the loss; per-microbatch losses are averaged so the result matches a
single-process model run on the full batch with a mean-reduced loss.
Activation shapes are exchanged once (first step) via object send/recv,
then P2P tensors flow with no per-step metadata.
"""

from typing import Callable, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


def split_into_stages(model: nn.Sequential, n_stages: int) -> List[nn.Sequential]:
    """Split an nn.Sequential into ``n_stages`` parameter-balanced chunks."""
    layers = list(model.children())
    weights = [max(sum(p.numel() for p in layer.parameters()), 1) for layer in layers]
    total = sum(weights)
    stages, current, acc = [], [], 0
    target = total / n_stages
    for layer, w in zip(layers, weights):
        current.append(layer)
        acc += w
        if acc >= target * (len(stages) + 1) and len(stages) < n_stages - 1:
            stages.append(nn.Sequential(*current))
            current = []
    stages.append(nn.Sequential(*current))
    while len(stages) < n_stages:  # degenerate tiny models
        stages.append(nn.Sequential())
    return stages


class PipelineParallelEngine:
    """One pipeline stage per rank; ``train_step`` runs a full GPipe cycle.

    Every rank constructs the engine with the SAME model (stages are
    selected locally) or with an explicit per-rank ``stage`` module.
    """

    def __init__(
        self,
        model: Optional[nn.Sequential] = None,
        stage: Optional[nn.Module] = None,
        num_microbatches: int = 4,
        group=None,
    ):
        if not dist.is_initialized():
            raise RuntimeError("PipelineParallelEngine needs torch.distributed initialized")
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        self.num_microbatches = num_microbatches
        if stage is not None:
            self.stage = stage
        elif model is not None:
            self.stage = split_into_stages(model, self.world)[self.rank]
        else:
            raise ValueError("pass either a Sequential model or this rank's stage")
        self.is_first = self.rank == 0
        self.is_last = self.rank == self.world - 1
        self._recv_shape = None   # learned on the first step (downstream)
        self._announced = False   # shape sent downstream exactly once

    # -- P2P helpers (global ranks; group kept for future sub-worlds) -----

    def _send(self, tensor, dst):
        dist.send(tensor.contiguous(), dst=dst)

    def _recv(self, shape, dtype):
        buf = torch.empty(shape, dtype=dtype)
        dist.recv(buf, src=self.rank - 1)
        return buf

    def _exchange_shape_once(self, example=None):
        """Downstream ranks learn their input shape/dtype from upstream."""
        if self.is_first or self._recv_shape is not None:
            return
        meta = [None]
        dist.recv_object_list(meta, src=self.rank - 1)
        self._recv_shape = meta[0]

    def _announce_shape(self, out):
        dist.send_object_list([(tuple(out.shape), out.dtype)], dst=self.rank + 1)

    # -- the GPipe cycle ---------------------------------------------------

    def train_step(
        self,
        inputs: Optional[torch.Tensor] = None,
        targets: Optional[torch.Tensor] = None,
        loss_fn: Optional[Callable] = None,
    ):
        """One optimizer-ready step: microbatched fwd + bwd.

        - rank 0 passes ``inputs`` (full batch, split on dim 0)
        - the LAST rank passes ``targets`` and ``loss_fn(output, target)``
        - returns the mean loss tensor on the last rank, else None
        Parameter ``.grad``s hold the full-batch gradients afterwards.
        """
        M = self.num_microbatches
        micro_in: List[torch.Tensor] = []
        micro_out: List[torch.Tensor] = []
        losses: List[torch.Tensor] = []

        if self.is_first:
            if inputs is None:
                raise ValueError("rank 0 must provide inputs")
            feeds = list(torch.chunk(inputs, M, dim=0))
        if self.is_last and (targets is None or loss_fn is None):
            raise ValueError("last rank must provide targets and loss_fn")
        if self.is_last:
            target_chunks = list(torch.chunk(targets, M, dim=0))

        # ---- forward wave ----
        for m in range(M):
            if self.is_first:
                x = feeds[m].detach()
            else:
                self._exchange_shape_once()
                x = self._recv(*self._recv_shape)
            x.requires_grad_(not self.is_first)
            y = self.stage(x)
            micro_in.append(x)
            micro_out.append(y)
            if not self.is_last:
                if not self._announced:
                    self._announce_shape(y)
                    self._announced = True
                self._send(y.detach(), self.rank + 1)

        # ---- backward wave (reverse microbatch order) ----
        for m in reversed(range(M)):
            y = micro_out[m]
            if self.is_last:
                loss = loss_fn(y, target_chunks[m]) / M  # mean over microbatches
                losses.append(loss.detach())
                loss.backward()
            else:
                gout = torch.empty_like(y)
                dist.recv(gout, src=self.rank + 1)
                y.backward(gout)
            if not self.is_first:
                self._send(micro_in[m].grad, self.rank - 1)

        if self.is_last:
            return torch.stack(losses).sum()
        return None

    def parameters(self):
        return self.stage.parameters()
