"""Expert parallelism (MoE) over RCCL all-to-all.

The reference has no EP dimension of its own — its only touchpoints are
DeepSpeed MoE leaf-module marking (reference accelerator.py:2287) and MoE
regional-compile flags (accelerator.py:1700-1707); SURVEY §2 lists the
all-to-all token-dispatch engine as the native component to build. This is
that engine, MI355X-shaped:

- **Dropless** routing: variable-split ``all_to_all_single`` carries exactly
  the tokens each expert receives — no capacity factor, no dropped tokens,
  no padding traffic. xGMI is point-to-point (7 links/GPU), so the dispatch
  volume per link is what matters, not NVSwitch-style uniform fan-out.
- Top-k softmax gate with the standard load-balance auxiliary loss
  (Switch/Mixtral form: E * sum_e f_e * P_e).
- Each rank owns ``n_experts / ep_world`` experts; expert weights are
  rank-LOCAL (marked ``_no_ddp_sync`` so the DDP engine neither broadcasts
  nor all-reduces them — combining DP over the gate/backbone with EP over
  the experts Just Works).
- gloo (CPU tests) lacks all_to_all: emulated with all_gather_object, same
  numerics, test-only path.
"""

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F


class _AllToAllVar(torch.autograd.Function):
    """Autograd-aware variable-split all-to-all on dim 0.

    backward = the same exchange with the split lists swapped.
    """

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.out_splits, ctx.in_splits, ctx.group = out_splits, in_splits, group
        return _all_to_all_var(x, out_splits, in_splits, group)

    @staticmethod
    def backward(ctx, grad):
        return (
            _AllToAllVar.apply(grad.contiguous(), ctx.in_splits, ctx.out_splits, ctx.group),
            None,
            None,
            None,
        )


def _all_to_all_var(x, out_splits, in_splits, group):
    """out_splits[r] rows are received FROM rank r; in_splits[r] rows of x go
    TO rank r."""
    if dist.get_backend(group) == "gloo":
        # emulation: every rank publishes its per-destination chunks
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        chunks = list(torch.split(x, in_splits, dim=0))
        gathered = [None] * world
        dist.all_gather_object(gathered, [c.detach().cpu() for c in chunks], group=group)
        mine = [gathered[r][rank].to(x.device, x.dtype) for r in range(world)]
        return torch.cat(mine, dim=0)
    out = x.new_empty((sum(out_splits),) + tuple(x.shape[1:]))
    dist.all_to_all_single(
        out, x.contiguous(), output_split_sizes=out_splits, input_split_sizes=in_splits, group=group
    )
    return out


class ExpertMLP(nn.Module):
    """SwiGLU expert (the Mixtral/Llama-MoE FFN shape)."""

    def __init__(self, hidden_size: int, intermediate_size: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden_size, intermediate_size, bias=False)
        self.up_proj = nn.Linear(hidden_size, intermediate_size, bias=False)
        self.down_proj = nn.Linear(intermediate_size, hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(F.silu(self.gate_proj(x)) * self.up_proj(x))


class ExpertParallelMoE(nn.Module):
    """Top-k MoE layer with experts sharded across the EP group.

    ``n_experts`` must divide by the EP world size; rank r owns experts
    [r*E_local, (r+1)*E_local). The gate is replicated (DP-synced); expert
    parameters carry ``_no_ddp_sync``.
    """

    def __init__(
        self,
        hidden_size: int,
        intermediate_size: int,
        n_experts: int,
        top_k: int = 2,
        group: Optional[dist.ProcessGroup] = None,
        aux_loss_coef: float = 0.01,
    ):
        super().__init__()
        self.hidden_size = hidden_size
        self.n_experts = n_experts
        self.top_k = top_k
        self.aux_loss_coef = aux_loss_coef
        self.group = group
        self.ep_world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.ep_rank = dist.get_rank(group) if dist.is_initialized() else 0
        if n_experts % self.ep_world != 0:
            raise ValueError(f"n_experts ({n_experts}) must divide by EP world ({self.ep_world})")
        self.local_experts = n_experts // self.ep_world
        self.gate = nn.Linear(hidden_size, n_experts, bias=False)
        self.experts = nn.ModuleList(
            ExpertMLP(hidden_size, intermediate_size) for _ in range(self.local_experts)
        )
        for p in self.experts.parameters():
            p._no_ddp_sync = True
        self.aux_loss = torch.zeros(())  # refreshed every forward

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # NOTE: dropless routing requires the host to know the split sizes
        # (.tolist() below) — one device sync per MoE layer per step. That is
        # inherent to variable-split all-to-all; capacity-padded routing
        # (fixed splits, graph-capturable) is the documented alternative.
        shape = x.shape
        x = x.reshape(-1, self.hidden_size)
        N = x.shape[0]
        logits = self.gate(x)  # [N, E]
        probs = F.softmax(logits.float(), dim=-1)
        top_p, top_i = probs.topk(self.top_k, dim=-1)          # [N, K]
        top_p = (top_p / top_p.sum(-1, keepdim=True)).to(x.dtype)

        # load-balance aux loss: E * sum_e (fraction routed to e) * (mean prob e)
        with torch.autocast(device_type=x.device.type, enabled=False):
            f_e = torch.zeros(self.n_experts, device=x.device, dtype=torch.float32)
            f_e.scatter_add_(
                0, top_i.reshape(-1), torch.full((N * self.top_k,), 1.0 / (N * self.top_k), device=x.device)
            )
        self.aux_loss = self.aux_loss_coef * self.n_experts * (f_e * probs.mean(0)).sum()

        # sort expanded token copies by destination expert
        flat_expert = top_i.reshape(-1)                        # [N*K]
        order = torch.argsort(flat_expert, stable=True)
        src_row = order // self.top_k                          # originating token
        send = x[src_row]                                      # [N*K, d]
        counts = torch.bincount(flat_expert, minlength=self.n_experts)  # per expert

        if self.ep_world > 1:
            per_rank = counts.reshape(self.ep_world, self.local_experts).sum(-1)
            in_splits = per_rank.tolist()
            # exchange per-expert counts so the receive side can segment
            all_counts = [torch.zeros_like(counts) for _ in range(self.ep_world)]
            dist.all_gather(all_counts, counts, group=self.group)
            lo = self.ep_rank * self.local_experts
            recv_per_rank = [int(c[lo : lo + self.local_experts].sum()) for c in all_counts]
            recv = _AllToAllVar.apply(send, recv_per_rank, in_splits, self.group)
            # received rows are ordered (src rank, expert); regroup per expert
            seg = torch.cat([c[lo : lo + self.local_experts] for c in all_counts]).reshape(
                self.ep_world, self.local_experts
            )
            by_expert_order = torch.argsort(
                torch.repeat_interleave(
                    torch.arange(self.ep_world * self.local_experts, device=x.device)
                    % self.local_experts,
                    seg.reshape(-1),
                ),
                stable=True,
            )
            grouped = recv[by_expert_order]
            expert_counts = seg.sum(0)                          # [local_experts]
        else:
            grouped = send
            expert_counts = counts

        # run local experts on their contiguous segments
        outs = []
        start = 0
        for e in range(self.local_experts):
            n_e = int(expert_counts[e])
            outs.append(self.experts[e](grouped[start : start + n_e]))
            start += n_e
        done = torch.cat(outs, dim=0) if outs else grouped

        if self.ep_world > 1:
            # undo the per-expert regroup, then the all-to-all
            inv = torch.empty_like(by_expert_order)
            inv[by_expert_order] = torch.arange(by_expert_order.numel(), device=x.device)
            done = done[inv]
            done = _AllToAllVar.apply(done, in_splits, recv_per_rank, self.group)

        # weighted scatter-add back to token order
        out = torch.zeros_like(x)
        w = top_p.reshape(-1)[order, None]
        out.index_add_(0, src_row, done * w)
        return out.reshape(shape)


def balance_loss(model: nn.Module) -> torch.Tensor:
    """Sum of all MoE aux losses in the model (add to the training loss)."""
    total = None
    for m in model.modules():
        if isinstance(m, ExpertParallelMoE):
            total = m.aux_loss if total is None else total + m.aux_loss
    if total is None:
        raise ValueError("model has no ExpertParallelMoE layers")
    return total
