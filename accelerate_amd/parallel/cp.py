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

The ring (P2P KV rotation) variant — `ring_attention` below — maps to
RCCL send/recv over the 7 xGMI links and keeps KV memory sequence-local,
so max sequence length scales ~n× past the all-gather variant.
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
    """Slice this rank's contiguous sequence shard (default dim 1 = [B, S]).

    The sequence length must divide the cp world size — a silent
    truncation of the tail would train on different data than the
    unsharded run (the reference's torch CP has the same requirement)."""
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world == 1:
        return tensor
    if tensor.shape[dim] % world != 0:
        raise ValueError(
            f"context parallelism requires the sequence length ({tensor.shape[dim]}, dim {dim}) "
            f"to divide cp_size ({world}); pad the batch to a multiple"
        )
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


# ---------------------------------------------------------------------------
# Ring (P2P KV rotation) context parallelism
# ---------------------------------------------------------------------------
# Reference parity: torch.distributed.tensor.experimental._attention's
# rotate_method='alltoall'/ring family. MI355X-first
# design: each step sends K/V one hop over a dedicated point-to-point xGMI
# link (7 links/GPU, ~153 GB/s each) while the local partial attention runs,
# so the gathered-KV memory never materializes — per-rank KV stays S/n.
# Partials are merged with the flash kernel's natural-log lse
# (out = Σ_i exp(lse_i - lse) · out_i); the backward ring re-uses the
# per-chunk flash backward with the GLOBAL merged lse (partial softmax rows
# are exactly exp(S_chunk - lse_global)) and rotates (k, v, dk, dv) so each
# chunk's accumulated dk/dv arrives home after a full revolution.


def _ring_shift(group, tensors):
    """Rotate each tensor one hop around the cp ring: send ours to rank+1,
    receive rank-1's. Recvs are posted before sends (both non-blocking) so
    the exchange cannot deadlock on either backend; message order pairs the
    tensors up."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if group is None:
        dst, src = (rank + 1) % world, (rank - 1) % world
    else:
        dst = dist.get_global_rank(group, (rank + 1) % world)
        src = dist.get_global_rank(group, (rank - 1) % world)
    outs, works = [], []
    for t in tensors:
        t = t.contiguous()
        buf = torch.empty_like(t)
        works.append(dist.irecv(buf, src=src, group=group))
        works.append(dist.isend(t, dst=dst, group=group))
        outs.append(buf)
    for w in works:
        w.wait()
    return outs


def _merge_partial(out, lse, o_i, l_i):
    """Fold one chunk's (normalized out_i, lse_i) into the running merged
    state (fp32 out, fp32 natural-log lse)."""
    if out is None:
        return o_i.float(), l_i.clone()
    new_lse = torch.logaddexp(lse, l_i)
    out = out * torch.exp(lse - new_lse).unsqueeze(-1) + o_i.float() * torch.exp(
        l_i - new_lse
    ).unsqueeze(-1)
    return out, new_lse


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal, scale):
        from ..ops.attention import _fwd_with_lse

        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        s = q.shape[2]
        k_cur, v_cur = k.contiguous(), v.contiguous()
        out, lse = None, None
        for step in range(world):
            c = (rank - step) % world  # absolute chunk index currently held
            offset = (rank - c) * s  # q_start relative to this chunk's keys
            nxt = _ring_shift(group, [k_cur, v_cur]) if step < world - 1 else None
            if (not causal) or offset >= 0:  # offset < 0: chunk entirely future
                with torch.no_grad():
                    o_i, l_i = _fwd_with_lse(q, k_cur, v_cur, causal, scale, offset)
                out, lse = _merge_partial(out, lse, o_i, l_i)
            if nxt is not None:
                k_cur, v_cur = nxt
        out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.meta = (group, causal, scale)
        return out

    @staticmethod
    def backward(ctx, dout):
        from ..ops.attention import _bwd_chunk

        q, k, v, out, lse = ctx.saved_tensors
        group, causal, scale = ctx.meta
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        s = q.shape[2]
        dout = dout.contiguous().to(q.dtype)
        k_cur, v_cur = k.contiguous(), v.contiguous()
        dk_cur = torch.zeros_like(k, dtype=torch.float32)
        dv_cur = torch.zeros_like(v, dtype=torch.float32)
        dq = torch.zeros_like(q, dtype=torch.float32)
        for step in range(world):
            c = (rank - step) % world
            offset = (rank - c) * s
            if (not causal) or offset >= 0:
                dq_i, dk_i, dv_i = _bwd_chunk(dout, q, k_cur, v_cur, out, lse, causal, scale, offset)
                dq += dq_i
                dk_cur += dk_i
                dv_cur += dv_i
            if step < world - 1:
                k_cur, v_cur, dk_cur, dv_cur = _ring_shift(group, [k_cur, v_cur, dk_cur, dv_cur])
        # after world-1 shifts we hold chunk (rank+1)%world's dk/dv: one more
        # hop delivers every chunk's accumulated gradient to its owner
        dk_cur, dv_cur = _ring_shift(group, [dk_cur, dv_cur])
        return dq.to(q.dtype), dk_cur.to(k.dtype), dv_cur.to(v.dtype), None, None, None


def ring_attention(q, k, v, group=None, causal=True, scale=None):
    """Ring-rotation context-parallel attention: q,k,v are this rank's
    sequence shard [B, H, S/n, D] (equal head counts — GQA is expanded by
    ``dispatch_attention`` before the ring). KV memory stays sequence-local
    on every rank, so max sequence scales ~n× past the all-gather variant."""
    import math as _math

    from ..ops.attention import flash_attention

    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if scale is None:
        scale = 1.0 / _math.sqrt(q.shape[-1])
    if world == 1:
        return flash_attention(q, k, v, causal=causal, scale=scale)
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return _RingAttention.apply(q, k, v, group, causal, scale)
