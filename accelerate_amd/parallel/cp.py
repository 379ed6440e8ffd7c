"""Context parallelism (sequence-sharded attention) over RCCL/xGMI
(reference: SURVEY.md §2.3 CP row, §5.7 — the reference delegates to torch's
experimental ``context_parallel`` with rotate method 'allgather' by default;
this is our own implementation of that default).

Every rank holds a contiguous sequence shard of the batch. Inside each
attention, K/V are all-gathered across the cp group (differentiable:
all-gather forward, reduce-scatter-style slice-sum backward) and the local
Q attends the full sequence with its absolute causal offset (our blockwise
flash attention's ``q_start``). Activations/grads outside attention stay
sequence-local, so per-GPU memory scales 1/n with the cp degree.

The ring (P2P KV rotation) variant maps to RCCL send/recv over the 7 xGMI
links and is the planned v2 upgrade of `_AllGatherSeq`.
"""

import torch
import torch.distributed as dist


class _AllGatherSeq(torch.autograd.Function):
    """All-gather on the sequence dim (dim 2 of [B,H,S,D]).

    Backward: dL/dx_local = Σ_ranks grad_full[our slice] — other ranks
    computed attention against OUR keys, so their gradient contributions for
    our shard live on their ranks. All-reduce the full gradient, keep our
    slice (reduce-scatter semantics; a ring/reduce_scatter_tensor variant is
    the RCCL-optimized v2)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        ctx.world = world
        if world == 1:
            return x
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        return torch.cat(parts, dim=2)

    @staticmethod
    def backward(ctx, grad):
        if ctx.world == 1:
            return grad, None
        grad = grad.contiguous()
        dist.all_reduce(grad, group=ctx.group)
        rank = dist.get_rank(ctx.group)
        s = grad.shape[2] // ctx.world
        return grad[:, :, rank * s : (rank + 1) * s].contiguous(), None


def context_parallel_attention(q, k, v, group=None, causal=True):
    """q,k,v: the LOCAL sequence shard [B, H, S/n, D]."""
    from ..ops.attention import flash_attention

    world = dist.get_world_size(group) if dist.is_initialized() else 1
    rank = dist.get_rank(group) if dist.is_initialized() else 0
    if world == 1:
        return flash_attention(q, k, v, causal=causal)
    k_full = _AllGatherSeq.apply(k, group)
    v_full = _AllGatherSeq.apply(v, group)
    s = q.shape[2]
    return flash_attention(q, k_full, v_full, causal=causal, q_start=rank * s)


def shard_sequence(tensor: torch.Tensor, group=None, dim: int = 1) -> torch.Tensor:
    """Slice this rank's contiguous sequence shard (default dim 1 = [B, S])."""
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return tensor
    rank = dist.get_rank(group)
    s = tensor.shape[dim] // world
    return tensor.narrow(dim, rank * s, s).contiguous()


def apply_context_parallel_llama(model, group=None):
    """Patch every LlamaAttention to gather KV across the cp group. The
    caller feeds each rank its sequence shard (``shard_sequence``) and the
    positional tables are offset per rank."""
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return model
    rank = dist.get_rank(group)

    for layer in model.layers:
        attn = layer.self_attn
        attn._cp_group = group

        def make_forward(a):
            import math as _math

            def forward(x, cos, sin, kv_cache=None):
                B, S, _ = x.shape
                q = a.q_proj(x).view(B, S, a.n_heads, a.head_dim).transpose(1, 2)
                k = a.k_proj(x).view(B, S, a.n_kv, a.head_dim).transpose(1, 2)
                v = a.v_proj(x).view(B, S, a.n_kv, a.head_dim).transpose(1, 2)
                from ..models.llama import apply_rope

                # absolute positions for this rank's shard
                q = apply_rope(q, cos[rank * S :], sin[rank * S :])
                k = apply_rope(k, cos[rank * S :], sin[rank * S :])
                if a.n_kv != a.n_heads:
                    rep = a.n_heads // a.n_kv
                    k = k.repeat_interleave(rep, dim=1)
                    v = v.repeat_interleave(rep, dim=1)
                ctx = context_parallel_attention(q, k, v, group=a._cp_group, causal=True)
                ctx = ctx.transpose(1, 2).reshape(B, S, -1)
                return a.o_proj(ctx)

            return forward

        attn.forward = make_forward(attn)
    return model
