"""Fused rotary embedding wrapper (CDNA4 kernel; backward = rotation by −θ)."""

import torch

from . import _load_extension


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        ext = _load_extension(required=True)
        ctx.save_for_backward(cos, sin)
        return ext.rope_bf16(x.contiguous(), cos, sin, False)

    @staticmethod
    def backward(ctx, dy):
        ext = _load_extension(required=True)
        cos, sin = ctx.saved_tensors
        return ext.rope_bf16(dy.contiguous(), cos, sin, True), None, None


def fused_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [B, H, S, D] bf16; cos/sin: [>=S, D/2] fp32 (position-offset applied
    by the caller via slicing)."""
    return _RopeFn.apply(x, cos.contiguous(), sin.contiguous())


def rope_is_fusable(x, cos) -> bool:
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.shape[-1] % 16 == 0
        and cos.dtype == torch.float32
        and cos.is_cuda
    )
