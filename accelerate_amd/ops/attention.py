"""Memory-efficient attention for MI355X training (flash-attention algorithm,
blockwise over Q/K with online softmax + logsumexp-recompute backward).

Replaces the O(S²)-materializing math path for long sequences: peak memory is
O(S·block) instead of O(S²) and every hot op is an MFMA-backed rocBLAS GEMM
batch. fp32 softmax statistics, bf16 GEMM operands. The API is stable so the
fully-fused CDNA4 HIP kernel can replace the torch-ops body without touching
models (`attn_impl="fused"`).

Forward saves only (q, k, v, out, logsumexp): backward recomputes P per
block (the flash-attention backward), with
  D  = rowsum(dout ⊙ out)
  P  = exp(q kᵀ·scale − L)
  dv = Pᵀ dout ;  dP = dout vᵀ ;  dS = P ⊙ (dP − D)·scale
  dq = dS k    ;  dk = dSᵀ q
"""

import math
import os
from typing import Optional

import torch


def _block(size, pref):
    return min(size, pref)


def _strides_ok(t) -> bool:
    # kernel reads views directly: innermost dim contiguous, 16 B alignment
    return t.stride(-1) == 1 and all(s % 8 == 0 for s in t.stride()[:-1])


def _fused_eligible(q, k, v) -> bool:
    """The CDNA4 kernel covers bf16, head_dim 64/128, Sq big enough that a
    128-row tile isn't pure padding, and GQA kv (Hkv divides Hq) — read
    zero-copy through strides. Everything else takes the torch path."""
    return (
        q.is_cuda
        and q.dtype == torch.bfloat16
        and q.shape[-1] in (64, 128)
        and q.shape[2] >= 32
        and q.shape[0] * q.shape[1] <= 65535
        and q.shape[1] % k.shape[1] == 0
        and _strides_ok(q)
        and _strides_ok(k)
        and _strides_ok(v)
    )


def _fwd_with_lse(q, k, v, causal, scale, past, q_block=1024, k_block=1024):
    """Per-chunk forward returning (normalized out, NATURAL-log lse).
    `past` = absolute query offset minus absolute key offset (query i
    attends keys <= past + i). Shared by the autograd Function and the
    ring-CP engine (parallel/ring.py)."""
    B, H, Sq, Dh = q.shape
    Sk = k.shape[2]

    if _fused_eligible(q, k, v):
        from . import _load_extension

        ext = _load_extension(required=True)  # GPU boxes must run native
        out, lse = ext.flash_attn_fwd(q, k, v, causal, scale, past)
        return out, lse

    # torch path computes with kv expanded to Hq heads
    if k.shape[1] != H:
        rep = H // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    out = torch.empty_like(q)
    lse = torch.empty(B, H, Sq, dtype=torch.float32, device=q.device)
    for q0 in range(0, Sq, q_block):
        q1 = min(q0 + q_block, Sq)
        qb = q[:, :, q0:q1]
        acc = torch.zeros(B, H, q1 - q0, Dh, dtype=torch.float32, device=q.device)
        m = torch.full((B, H, q1 - q0), -float("inf"), dtype=torch.float32, device=q.device)
        denom = torch.zeros(B, H, q1 - q0, dtype=torch.float32, device=q.device)
        k_hi = Sk if not causal else min(Sk, past + q1)
        for k0 in range(0, k_hi, k_block):
            k1 = min(k0 + k_block, k_hi)
            s = torch.matmul(qb, k[:, :, k0:k1].transpose(-1, -2)).float() * scale
            if causal and k1 > past + q0:
                qi = torch.arange(q0, q1, device=q.device)[:, None]
                ki = torch.arange(k0, k1, device=q.device)[None, :]
                s = s.masked_fill(ki > past + qi, -float("inf"))
            blk_max = s.amax(-1)
            new_m = torch.maximum(m, blk_max)
            corr = torch.exp(m - new_m)
            p = torch.exp(s - new_m[..., None])
            acc = acc * corr[..., None] + torch.matmul(p.to(v.dtype), v[:, :, k0:k1]).float()
            denom = denom * corr + p.sum(-1)
            m = new_m
        out[:, :, q0:q1] = (acc / denom[..., None].clamp_min(1e-30)).to(q.dtype)
        lse[:, :, q0:q1] = m + denom.clamp_min(1e-30).log()
    return out, lse


def _bwd_chunk(dout, q, k, v, out, lse, causal, scale, past, q_block=1024, k_block=1024):
    """Per-chunk backward given the GLOBAL (merged) out/lse — returns
    (dq_partial fp32, dk_chunk fp32, dv_chunk fp32). With a global lse the
    chunk probabilities are exp(S - lse): partial rows of the full softmax,
    which is exactly what ring attention accumulates per rotation step."""
    B, H, Sq, Dh = q.shape
    Hkv = k.shape[1]
    rep = H // Hkv

    if _fused_eligible(q, k, v) and os.environ.get("ACCELERATE_AMD_FA_BWD", "1") == "1":
        from . import _load_extension

        ext = _load_extension(required=True)
        dq, dk, dv = ext.flash_attn_bwd(dout, q, k, v, out, lse, causal, scale, past)
        if rep > 1:
            dk = dk.view(B, Hkv, rep, *dk.shape[2:]).float().sum(2)
            dv = dv.view(B, Hkv, rep, *dv.shape[2:]).float().sum(2)
        return dq.float(), dk.float(), dv.float()
    if rep > 1:
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    Sk = k.shape[2]
    dq = torch.zeros_like(q, dtype=torch.float32)
    dk = torch.zeros_like(k, dtype=torch.float32)
    dv = torch.zeros_like(v, dtype=torch.float32)
    Drow = (dout.float() * out.float()).sum(-1)  # [B,H,Sq]
    for q0 in range(0, Sq, q_block):
        q1 = min(q0 + q_block, Sq)
        qb = q[:, :, q0:q1]
        dob = dout[:, :, q0:q1]
        Lb = lse[:, :, q0:q1]
        Db = Drow[:, :, q0:q1]
        k_hi = Sk if not causal else min(Sk, past + q1)
        for k0 in range(0, k_hi, k_block):
            k1 = min(k0 + k_block, k_hi)
            kb, vb = k[:, :, k0:k1], v[:, :, k0:k1]
            s = torch.matmul(qb, kb.transpose(-1, -2)).float() * scale
            if causal and k1 > past + q0:
                qi = torch.arange(q0, q1, device=q.device)[:, None]
                ki = torch.arange(k0, k1, device=q.device)[None, :]
                s = s.masked_fill(ki > past + qi, -float("inf"))
            p = torch.exp(s - Lb[..., None])
            pb = p.to(q.dtype)
            dv[:, :, k0:k1] += torch.matmul(pb.transpose(-1, -2), dob).float()
            dp = torch.matmul(dob, vb.transpose(-1, -2)).float()
            ds = (p * (dp - Db[..., None]) * scale).to(q.dtype)
            dq[:, :, q0:q1] += torch.matmul(ds, kb).float()
            dk[:, :, k0:k1] += torch.matmul(ds.transpose(-1, -2), qb).float()
    if rep > 1:
        dk = dk.view(B, Hkv, rep, Sk, Dh).sum(2)
        dv = dv.view(B, Hkv, rep, Sk, Dh).sum(2)
    return dq, dk, dv


class _FlashAttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, q_block, k_block, q_start=None):
        # q,k,v: [B, H, S, D] (kv may have S_k != S_q for cache decode)
        B, H, Sq, Dh = q.shape
        Sk = k.shape[2]
        # causal offset: query i attends keys <= past + i. Default assumes the
        # queries are the LAST Sq positions (kv-cache decode); context
        # parallelism passes the rank's absolute start explicitly.
        past = (Sk - Sq) if q_start is None else q_start

        if _fused_eligible(q, k, v):
            from . import _load_extension

            ext = _load_extension(required=True)  # GPU boxes must run native
            out, lse = ext.flash_attn_fwd(q, k, v, causal, scale, past)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.meta = (causal, scale, q_block, k_block, past)
            return out

        # torch path computes with kv expanded to Hq heads; the UNexpanded
        # tensors are saved (backward re-expands and group-sums dk/dv)
        k_orig, v_orig = k, v
        if k.shape[1] != H:
            rep = H // k.shape[1]
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        out = torch.empty_like(q)
        lse = torch.empty(B, H, Sq, dtype=torch.float32, device=q.device)

        for q0 in range(0, Sq, q_block):
            q1 = min(q0 + q_block, Sq)
            qb = q[:, :, q0:q1]
            acc = torch.zeros(B, H, q1 - q0, Dh, dtype=torch.float32, device=q.device)
            m = torch.full((B, H, q1 - q0), -float("inf"), dtype=torch.float32, device=q.device)
            denom = torch.zeros(B, H, q1 - q0, dtype=torch.float32, device=q.device)
            k_hi = Sk if not causal else min(Sk, past + q1)
            for k0 in range(0, k_hi, k_block):
                k1 = min(k0 + k_block, k_hi)
                s = torch.matmul(qb, k[:, :, k0:k1].transpose(-1, -2)).float() * scale
                if causal and k1 > past + q0:
                    qi = torch.arange(q0, q1, device=q.device)[:, None]
                    ki = torch.arange(k0, k1, device=q.device)[None, :]
                    s = s.masked_fill(ki > past + qi, -float("inf"))
                blk_max = s.amax(-1)
                new_m = torch.maximum(m, blk_max)
                # NOTE: every causal row sees key 0, so new_m is finite for all
                # processed rows — no nan_to_num pass needed (0.5 GB sweep saved)
                # rows still at -inf (fully masked so far) contribute nothing
                corr = torch.exp(m - new_m)
                p = torch.exp(s - new_m[..., None])
                acc = acc * corr[..., None] + torch.matmul(p.to(v.dtype), v[:, :, k0:k1]).float()
                denom = denom * corr + p.sum(-1)
                m = new_m
            out[:, :, q0:q1] = (acc / denom[..., None].clamp_min(1e-30)).to(q.dtype)
            lse[:, :, q0:q1] = m + denom.clamp_min(1e-30).log()

        ctx.save_for_backward(q, k_orig, v_orig, out, lse)
        ctx.meta = (causal, scale, q_block, k_block, past)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        causal, scale, q_block, k_block, past = ctx.meta
        B, H, Sq, Dh = q.shape
        Hkv = k.shape[1]
        rep = H // Hkv

        if _fused_eligible(q, k, v) and os.environ.get("ACCELERATE_AMD_FA_BWD", "1") == "1":
            from . import _load_extension

            ext = _load_extension(required=True)
            dq, dk, dv = ext.flash_attn_bwd(dout, q, k, v, out, lse, causal, scale, past)
            if rep > 1:  # sum the per-q-head partials down to the kv heads
                dk = dk.view(B, Hkv, rep, *dk.shape[2:]).float().sum(2).to(k.dtype)
                dv = dv.view(B, Hkv, rep, *dv.shape[2:]).float().sum(2).to(v.dtype)
            return dq, dk, dv, None, None, None, None, None
        if rep > 1:
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        Sk = k.shape[2]
        dq = torch.zeros_like(q, dtype=torch.float32)
        dk = torch.zeros_like(k, dtype=torch.float32)
        dv = torch.zeros_like(v, dtype=torch.float32)
        Drow = (dout.float() * out.float()).sum(-1)  # [B,H,Sq]

        for q0 in range(0, Sq, q_block):
            q1 = min(q0 + q_block, Sq)
            qb = q[:, :, q0:q1]
            dob = dout[:, :, q0:q1]
            Lb = lse[:, :, q0:q1]
            Db = Drow[:, :, q0:q1]
            k_hi = Sk if not causal else min(Sk, past + q1)
            for k0 in range(0, k_hi, k_block):
                k1 = min(k0 + k_block, k_hi)
                kb, vb = k[:, :, k0:k1], v[:, :, k0:k1]
                s = torch.matmul(qb, kb.transpose(-1, -2)).float() * scale
                if causal and k1 > past + q0:
                    qi = torch.arange(q0, q1, device=q.device)[:, None]
                    ki = torch.arange(k0, k1, device=q.device)[None, :]
                    s = s.masked_fill(ki > past + qi, -float("inf"))
                p = torch.exp(s - Lb[..., None])
                pb = p.to(q.dtype)
                dv[:, :, k0:k1] += torch.matmul(pb.transpose(-1, -2), dob).float()
                dp = torch.matmul(dob, vb.transpose(-1, -2)).float()
                ds = (p * (dp - Db[..., None]) * scale).to(q.dtype)
                dq[:, :, q0:q1] += torch.matmul(ds, kb).float()
                dk[:, :, k0:k1] += torch.matmul(ds.transpose(-1, -2), qb).float()

        if rep > 1:
            dk = dk.view(B, Hkv, rep, Sk, Dh).sum(2)
            dv = dv.view(B, Hkv, rep, Sk, Dh).sum(2)
        return dq.to(q.dtype), dk.to(q.dtype), dv.to(q.dtype), None, None, None, None, None


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
    q_block: int = 1024,
    k_block: int = 1024,
    q_start: Optional[int] = None,
) -> torch.Tensor:
    """Blockwise attention over [B, H, S, D] tensors. ``k``/``v`` may carry
    FEWER heads (GQA): any Hkv dividing H is expanded in-kernel on GPU
    (zero-copy through strides, no repeat_interleave) or by the torch path
    on CPU. ``q_start`` is the absolute position of q[...,0] for causal
    masking (context parallelism)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _FlashAttentionFn.apply(
        q, k, v, causal, scale, _block(q.shape[2], q_block), _block(k.shape[2], k_block), q_start
    )


# alias used by models (the fused HIP kernel will take over this symbol)
flash_attention_forward = flash_attention


# ---------------------------------------------------------------------------
# Sequence-parallel dispatch (the model-generic CP/SP route)
#
# `Accelerator.prepare` registers the active context here when
# ParallelismConfig declares cp_size > 1; any model whose attention goes
# through `dispatch_attention` (all in-repo families) is then context- or
# Ulysses-parallel with NO model-specific patching (reference contrast:
# accelerator.py:1658 _prepare_cp requires torch experimental
# context_parallel + SDPA hooks).
# ---------------------------------------------------------------------------

_SEQ_PARALLEL = {"mode": None, "group": None}


def set_sequence_parallel(mode: Optional[str], group=None):
    """mode: None (off) | 'allgather' (CP: KV all-gather) | 'ring' (CP:
    P2P KV rotation, sequence-local memory) | 'ulysses' (SP: dual
    all-to-all head resharding)."""
    if mode not in (None, "allgather", "ring", "ulysses"):
        raise ValueError(f"unknown sequence-parallel mode {mode!r}")
    _SEQ_PARALLEL["mode"] = mode
    _SEQ_PARALLEL["group"] = group


def sequence_parallel_info():
    """(mode, group, rank, world) of the active sequence-parallel context;
    (None, None, 0, 1) when off."""
    import torch.distributed as dist

    mode, group = _SEQ_PARALLEL["mode"], _SEQ_PARALLEL["group"]
    if mode is None or not dist.is_initialized():
        return None, None, 0, 1
    world = dist.get_world_size(group)
    if world == 1:
        return None, None, 0, 1
    return mode, group, dist.get_rank(group), world


def dispatch_attention(q, k, v, causal: bool = True):
    """Attention entry point for models: local flash attention normally;
    under an active sequence-parallel context, q/k/v are this rank's
    sequence shard and the collective pattern of the registered mode runs
    around the kernel. GQA inputs are head-expanded first on the parallel
    paths (the collectives need matching head counts)."""
    mode, group, _, _ = sequence_parallel_info()
    if mode is None:
        return flash_attention(q, k, v, causal=causal)
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    if mode == "allgather":
        from ..parallel.cp import context_parallel_attention

        return context_parallel_attention(q, k, v, group=group, causal=causal)
    if mode == "ring":
        from ..parallel.cp import ring_attention

        return ring_attention(q, k, v, group=group, causal=causal)
    from ..parallel.sp import ulysses_attention

    return ulysses_attention(q, k, v, group=group, causal=causal)
