"""Tensor parallelism over RCCL/xGMI (reference: SURVEY.md §2.3 TP row,
§2.9 N14 — the reference delegates to torch DTensor TP; this is our own
megatron-style column/row-parallel implementation).

- ColumnParallelLinear: weight sharded on the OUTPUT dim; input is
  replicated (identity forward / grad all-reduce backward); output is the
  local shard (or all-gathered with ``gather_output=True``).
- RowParallelLinear: weight sharded on the INPUT dim; input is the local
  shard; partial outputs are all-reduced (identity backward).
- ``tp_parallelize_llama`` shards a LlamaForCausalLM in place: q/k/v/gate/up
  column-parallel, o/down row-parallel, head counts divided per rank —
  attention then runs entirely on local heads with ONE all-reduce per
  attention block and one per MLP (the megatron pattern; over xGMI these are
  the only TP collectives per layer).
"""

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient (input is replicated)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if dist.is_initialized() and dist.get_world_size(ctx.group) > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward (sum partials); identity backward."""

    @staticmethod
    def forward(ctx, x, group):
        if dist.is_initialized() and dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _GatherFromTP(torch.autograd.Function):
    """All-gather shards on the last dim; slice gradient backward."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        ctx.world = world
        if world == 1:
            return x
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, grad):
        if ctx.world == 1:
            return grad, None
        rank = dist.get_rank(ctx.group)
        size = grad.shape[-1] // ctx.world
        return grad[..., rank * size : (rank + 1) * size].contiguous(), None


class ColumnParallelLinear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, gather_output=False, group=None, dtype=None):
        super().__init__()
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        assert out_features % world == 0, "out_features must divide tp world"
        self.group = group
        self.gather_output = gather_output
        self.out_per_rank = out_features // world
        self.weight = nn.Parameter(torch.empty(self.out_per_rank, in_features, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.out_per_rank, dtype=dtype)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=5**0.5)

    def forward(self, x):
        x = _CopyToTP.apply(x, self.group)
        y = nn.functional.linear(x, self.weight, self.bias)
        if self.gather_output:
            y = _GatherFromTP.apply(y, self.group)
        return y

    @classmethod
    def from_linear(cls, linear: nn.Linear, group=None, gather_output=False):
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        rank = dist.get_rank(group) if dist.is_initialized() else 0
        mod = cls.__new__(cls)
        nn.Module.__init__(mod)
        mod.group = group
        mod.gather_output = gather_output
        mod.out_per_rank = linear.out_features // world
        lo = rank * mod.out_per_rank
        mod.weight = nn.Parameter(linear.weight[lo : lo + mod.out_per_rank].detach().clone())
        mod.bias = (
            nn.Parameter(linear.bias[lo : lo + mod.out_per_rank].detach().clone()) if linear.bias is not None else None
        )
        return mod


class RowParallelLinear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, input_is_parallel=True, group=None, dtype=None):
        super().__init__()
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        assert in_features % world == 0, "in_features must divide tp world"
        self.group = group
        self.input_is_parallel = input_is_parallel
        self.in_per_rank = in_features // world
        self.weight = nn.Parameter(torch.empty(out_features, self.in_per_rank, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=5**0.5)

    def forward(self, x):
        if not self.input_is_parallel:
            world = dist.get_world_size(self.group) if dist.is_initialized() else 1
            rank = dist.get_rank(self.group) if dist.is_initialized() else 0
            x = x[..., rank * self.in_per_rank : (rank + 1) * self.in_per_rank]
        y = nn.functional.linear(x, self.weight)
        y = _ReduceFromTP.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y

    @classmethod
    def from_linear(cls, linear: nn.Linear, group=None, input_is_parallel=True):
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        rank = dist.get_rank(group) if dist.is_initialized() else 0
        mod = cls.__new__(cls)
        nn.Module.__init__(mod)
        mod.group = group
        mod.input_is_parallel = input_is_parallel
        mod.in_per_rank = linear.in_features // world
        lo = rank * mod.in_per_rank
        mod.weight = nn.Parameter(linear.weight[:, lo : lo + mod.in_per_rank].detach().clone())
        mod.bias = nn.Parameter(linear.bias.detach().clone()) if linear.bias is not None else None
        return mod


def tp_parallelize_llama(model, group=None):
    """Shard a LlamaForCausalLM in place (megatron pattern): column q/k/v +
    gate/up, row o/down; per-rank head counts. One all-reduce per attention
    and one per MLP per layer."""
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return model
    for layer in model.layers:
        attn = layer.self_attn
        assert attn.n_heads % world == 0 and attn.n_kv % world == 0, "heads must divide tp world"
        attn.q_proj = ColumnParallelLinear.from_linear(attn.q_proj, group)
        attn.k_proj = ColumnParallelLinear.from_linear(attn.k_proj, group)
        attn.v_proj = ColumnParallelLinear.from_linear(attn.v_proj, group)
        attn.o_proj = RowParallelLinear.from_linear(attn.o_proj, group)
        attn.n_heads //= world
        attn.n_kv //= world
        mlp = layer.mlp
        mlp.gate_proj = ColumnParallelLinear.from_linear(mlp.gate_proj, group)
        mlp.up_proj = ColumnParallelLinear.from_linear(mlp.up_proj, group)
        mlp.down_proj = RowParallelLinear.from_linear(mlp.down_proj, group)
    return model
