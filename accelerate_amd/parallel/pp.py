"""Pipeline-parallel TRAINING (GPipe and 1F1B schedules over P2P).

The reference implements pipeline parallelism for inference only and
explicitly raises for training (reference accelerator.py:795-799
NotImplementedError; inference lives in inference.py). This engine goes
beyond it: microbatched forward/backward with activation hand-off over
point-to-point sends — xGMI-adjacent on RCCL, TCP on gloo (how the
2-process CPU oracles run).

Schedules (both produce IDENTICAL gradients; they differ in peak memory):

- ``gpipe``  — all forwards, then all backwards (reversed order). Peak
  resident activations = num_microbatches.
- ``1f1b``   — stage r runs (S-1-r) warmup forwards, then alternates one
  forward / one backward, then drains backwards. Peak resident
  activations = S - r: INDEPENDENT of the microbatch count, which is what
  lets deep pipelines use many microbatches to hide bubble time.

Sends are ``isend`` (buffered, waited at step end) so the blocking-recv
orderings of adjacent ranks can never form a cycle; per-channel message
order is monotonic in microbatch index for both schedules.

Gradients accumulate across microbatches in the stage's parameters; the
caller steps its optimizer once per ``train_step``. The LAST stage owns
the loss; per-microbatch losses are scaled 1/M so the result matches a
single-process model with a mean-reduced loss. Activation shape/dtype is
announced downstream exactly once.
"""

from typing import Callable, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


def split_into_stages(model: nn.Sequential, n_stages: int) -> List[nn.Sequential]:
    """Split an nn.Sequential into ``n_stages`` parameter-balanced chunks.

    Each cut lands at the layer boundary whose parameter prefix-sum is
    CLOSEST to the ideal k/n split (a forward greedy cut can strand a
    parameter-less activation layer as a whole stage — e.g. 3-way
    splitting [Linear, Tanh, Linear, Tanh, Linear] used to produce a
    [Tanh]-only stage, which breaks per-stage optimizers)."""
    layers = list(model.children())
    weights = [max(sum(p.numel() for p in layer.parameters()), 1) for layer in layers]
    total = sum(weights)
    prefix = []
    acc = 0
    for w in weights:
        acc += w
        prefix.append(acc)
    cuts = []
    prev = 0
    for k in range(1, n_stages):
        ideal = total * k / n_stages
        lo = prev + 1  # at least one layer per stage
        hi = len(layers) - (n_stages - k)  # leave one layer per later stage
        if lo > hi:
            break
        best = min(range(lo, hi + 1), key=lambda i: abs(prefix[i - 1] - ideal))
        cuts.append(best)
        prev = best
    stages = []
    bounds = [0] + cuts + [len(layers)]
    for a, b in zip(bounds[:-1], bounds[1:]):
        stages.append(nn.Sequential(*layers[a:b]))
    while len(stages) < n_stages:  # degenerate tiny models
        stages.append(nn.Sequential())
    return stages


class PipelineParallelEngine:
    """One pipeline stage per rank; ``train_step`` runs a full cycle.

    Every rank constructs the engine with the SAME model (stages are
    selected locally) or with an explicit per-rank ``stage`` module.
    """

    def __init__(
        self,
        model: Optional[nn.Sequential] = None,
        stage: Optional[nn.Module] = None,
        num_microbatches: int = 4,
        schedule: str = "gpipe",
        group=None,
        device: Optional[torch.device] = None,
    ):
        if not dist.is_initialized():
            raise RuntimeError("PipelineParallelEngine needs torch.distributed initialized")
        if schedule not in ("gpipe", "1f1b"):
            raise ValueError(f"unknown schedule {schedule!r} (gpipe | 1f1b)")
        self.group = group
        self.world = dist.get_world_size(group)
        self.rank = dist.get_rank(group)
        self.num_microbatches = num_microbatches
        self.schedule = schedule
        if stage is not None:
            self.stage = stage
        elif model is not None:
            self.stage = split_into_stages(model, self.world)[self.rank]
        else:
            raise ValueError("pass either a Sequential model or this rank's stage")
        # Under an RCCL world the stage and every P2P buffer must live on this
        # rank's GPU (recv into a CPU tensor fails on the nccl backend); under
        # gloo everything stays on CPU. Resolve from the caller, the current
        # PartialState device, or the stage's own parameters — in that order.
        if device is None:
            from ..state import PartialState

            state = PartialState._shared_state
            if state.get("device") is not None and dist.get_backend() != "gloo":
                device = state["device"]
            else:
                device = next(self.stage.parameters(), torch.empty(0)).device
        self.device = torch.device(device)
        self.stage.to(self.device)
        self.is_first = self.rank == 0
        self.is_last = self.rank == self.world - 1
        self._recv_shape = None   # learned on the first step (downstream)
        self._announced = False   # shape sent downstream exactly once
        self._inflight = []       # (work, tensor) pairs kept alive until wait

    # -- P2P plumbing ------------------------------------------------------

    def _send_async(self, tensor, dst):
        t = tensor.contiguous()
        self._inflight.append((dist.isend(t, dst=dst), t))

    def _drain_sends(self):
        for work, _ in self._inflight:
            work.wait()
        self._inflight = []

    def _recv_from_prev(self):
        shape, dtype = self._recv_shape
        buf = torch.empty(shape, dtype=dtype, device=self.device)
        dist.recv(buf, src=self.rank - 1)
        return buf

    def _learn_shape_once(self):
        if self.is_first or self._recv_shape is not None:
            return
        meta = [None]
        dist.recv_object_list(meta, src=self.rank - 1)
        self._recv_shape = meta[0]

    def _announce_shape_once(self, out):
        if self.is_last or self._announced:
            return
        dist.send_object_list([(tuple(out.shape), out.dtype)], dst=self.rank + 1)
        self._announced = True

    # -- per-microbatch halves --------------------------------------------

    def _forward_one(self, m, ctx):
        if self.is_first:
            x = ctx["feeds"][m].detach()
        else:
            self._learn_shape_once()
            x = self._recv_from_prev()
        x.requires_grad_(not self.is_first)
        y = self.stage(x)
        ctx["x"][m], ctx["y"][m] = x, y
        if not self.is_last:
            self._announce_shape_once(y)
            self._send_async(y.detach(), self.rank + 1)

    def _backward_one(self, m, ctx):
        y = ctx["y"][m]
        if self.is_last:
            loss = ctx["loss_fn"](y, ctx["targets"][m]) / self.num_microbatches
            ctx["losses"].append(loss.detach())
            loss.backward()
        else:
            gout = torch.empty_like(y)
            dist.recv(gout, src=self.rank + 1)
            y.backward(gout)
        if not self.is_first:
            self._send_async(ctx["x"][m].grad, self.rank - 1)
        ctx["x"][m] = ctx["y"][m] = None  # release the microbatch's memory

    # -- the training step -------------------------------------------------

    def train_step(
        self,
        inputs: Optional[torch.Tensor] = None,
        targets: Optional[torch.Tensor] = None,
        loss_fn: Optional[Callable] = None,
    ):
        """One optimizer-ready cycle: M microbatched forwards + backwards.

        - rank 0 passes ``inputs`` (full batch, chunked on dim 0)
        - the LAST rank passes ``targets`` and ``loss_fn(output, target)``
        - returns the mean loss tensor on the last rank, else None
        Parameter ``.grad``s hold the full-batch gradients afterwards.
        """
        M = self.num_microbatches
        ctx = {"x": [None] * M, "y": [None] * M, "losses": [], "loss_fn": loss_fn}
        if self.is_first:
            if inputs is None:
                raise ValueError("rank 0 must provide inputs")
            ctx["feeds"] = list(torch.chunk(inputs.to(self.device), M, dim=0))
        if self.is_last:
            if targets is None or loss_fn is None:
                raise ValueError("last rank must provide targets and loss_fn")
            ctx["targets"] = list(torch.chunk(targets.to(self.device), M, dim=0))

        if self.schedule == "gpipe":
            for m in range(M):
                self._forward_one(m, ctx)
            for m in reversed(range(M)):
                self._backward_one(m, ctx)
        else:  # 1f1b
            warmup = min(self.world - 1 - self.rank, M)
            fwd = bwd = 0
            for _ in range(warmup):
                self._forward_one(fwd, ctx)
                fwd += 1
            while fwd < M:
                self._forward_one(fwd, ctx)
                fwd += 1
                self._backward_one(bwd, ctx)
                bwd += 1
            while bwd < M:
                self._backward_one(bwd, ctx)
                bwd += 1

        self._drain_sends()
        if self.is_last:
            return torch.stack(ctx["losses"]).sum()
        return None

    def parameters(self):
        return self.stage.parameters()
