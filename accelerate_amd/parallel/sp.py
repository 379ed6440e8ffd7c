"""Ulysses-style sequence parallelism: two all-to-alls around attention
(reference: SURVEY.md §2.3 SP row, §5.7 — the reference delegates to
DeepSpeed ALST/UlyssesSP; this is our own RCCL all-to-all implementation).

Activations are sequence-sharded everywhere except inside attention: the
first all-to-all reshards [B, H, S/n, D] → [B, H/n, S, D] (full sequence,
local head group), attention runs exactly as usual on n-times-fewer heads,
and the second all-to-all reshards back. Head count must divide sp world.
Over xGMI an all-to-all is n·(n-1) point-to-point transfers — bandwidth-
optimal on the 7-link topology (each link carries 1/n of the payload).
"""

import torch
import torch.distributed as dist


class _AllToAllHeadsSeq(torch.autograd.Function):
    """[B, H, S_local, D] → [B, H/n, S_full, D] (seq-shards → head-shards).
    Backward is the inverse resharding of the gradient."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _a2a_scatter_heads(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _a2a_gather_heads(grad, ctx.group), None


class _AllToAllSeqHeads(torch.autograd.Function):
    """[B, H/n, S_full, D] → [B, H, S_local, D] (head-shards → seq-shards)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _a2a_gather_heads(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _a2a_scatter_heads(grad, ctx.group), None


def _all_to_all(send, group):
    """RCCL all_to_all; gloo (CPU tests) lacks alltoall → all_gather emulation."""
    world = len(send)
    recv = [torch.empty_like(send[0]) for _ in range(world)]
    if dist.get_backend(group) == "nccl":
        dist.all_to_all(recv, [c.contiguous() for c in send], group=group)
        return recv
    me = dist.get_rank(group)
    stacked = torch.stack([c.contiguous() for c in send])
    gathered = [torch.empty_like(stacked) for _ in range(world)]
    dist.all_gather(gathered, stacked, group=group)
    return [gathered[j][me] for j in range(world)]


def _a2a_scatter_heads(x, group):
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return x
    B, H, S, D = x.shape
    assert H % world == 0, "head count must divide sp world size"
    recv = _all_to_all(list(x.chunk(world, dim=1)), group)  # head group j -> rank j
    return torch.cat(recv, dim=2)  # our head group, sequence re-assembled


def _a2a_gather_heads(x, group):
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return x
    recv = _all_to_all(list(x.chunk(world, dim=2)), group)  # seq shard j -> rank j
    return torch.cat(recv, dim=1)  # full heads, our sequence shard


def ulysses_attention(q, k, v, group=None, causal=True):
    """q,k,v: local sequence shard [B, H, S/n, D] with FULL head count."""
    from ..ops.attention import flash_attention

    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return flash_attention(q, k, v, causal=causal)
    qh = _AllToAllHeadsSeq.apply(q, group)
    kh = _AllToAllHeadsSeq.apply(k, group)
    vh = _AllToAllHeadsSeq.apply(v, group)
    out = flash_attention(qh, kh, vh, causal=causal)
    return _AllToAllSeqHeads.apply(out, group)


def apply_ulysses_llama(model, group=None):
    """Patch every LlamaAttention for Ulysses SP (same input contract as CP:
    each rank feeds its sequence shard; RoPE offset per rank)."""
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return model
    rank = dist.get_rank(group)

    for layer in model.layers:
        attn = layer.self_attn

        def make_forward(a):
            def forward(x, cos, sin, kv_cache=None):
                B, S, _ = x.shape
                q = a.q_proj(x).view(B, S, a.n_heads, a.head_dim).transpose(1, 2)
                k = a.k_proj(x).view(B, S, a.n_kv, a.head_dim).transpose(1, 2)
                v = a.v_proj(x).view(B, S, a.n_kv, a.head_dim).transpose(1, 2)
                from ..models.llama import apply_rope

                q = apply_rope(q, cos[rank * S :], sin[rank * S :])
                k = apply_rope(k, cos[rank * S :], sin[rank * S :])
                if a.n_kv != a.n_heads:
                    rep = a.n_heads // a.n_kv
                    k = k.repeat_interleave(rep, dim=1)
                    v = v.repeat_interleave(rep, dim=1)
                ctx = ulysses_attention(q, k, v, group=group, causal=True)
                ctx = ctx.transpose(1, 2).reshape(B, S, -1)
                return a.o_proj(ctx)

            return forward

        attn.forward = make_forward(attn)
    return model
