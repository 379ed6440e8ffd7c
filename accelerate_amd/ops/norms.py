"""Fused CDNA4 normalization modules (bf16 in/out, fp32 statistics).

Drop-in replacements for nn.LayerNorm / Llama RMSNorm on MI355X: one fused
kernel forward, two backward (vs torch's fp32 path with cast copies around
every call — the bench profile showed LN + casts ≈ 6% of the BERT step).
CPU or non-bf16 inputs fall back to the eager math so the modules stay
correct everywhere.
"""

import os

import torch
import torch.nn as nn

from . import _load_extension


class _FusedLayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = _load_extension(required=True)
        y, mean, rstd = ext.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load_extension(required=True)
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.layernorm_bwd(dy, x, weight, mean, rstd)
        return dx, dw, db, None


class _FusedRMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = _load_extension(required=True)
        y, rstd = ext.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load_extension(required=True)
        x, weight, rstd = ctx.saved_tensors
        dx, dw = ext.rmsnorm_bwd(dy, x, weight, rstd)
        return dx, dw, None


def _use_fused(x, *params):
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.shape[-1] % 8 == 0
        and all(p.dtype == torch.bfloat16 for p in params)
    )


class FusedLayerNorm(nn.LayerNorm):
    """nn.LayerNorm with the fused bf16 CDNA4 kernel on the hot path."""

    def forward(self, x):
        if _use_fused(x, self.weight, self.bias) and len(self.normalized_shape) == 1 and x.shape[-1] <= 2048:
            return _FusedLayerNormFn.apply(x.contiguous(), self.weight, self.bias, self.eps)
        return super().forward(x)


class FusedRMSNorm(nn.Module):
    def __init__(self, hidden_size, eps=1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        if _use_fused(x, self.weight) and x.shape[-1] <= 8192:
            return _FusedRMSNormFn.apply(x.contiguous(), self.weight, self.eps)
        dtype = x.dtype
        xf = x.float()
        xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return self.weight.to(dtype) * xf.to(dtype)


def convert_to_fused_norms(model: nn.Module) -> nn.Module:
    """Swap nn.LayerNorm (1-D) / RMSNorm-like modules for the fused versions."""
    for name, module in model.named_modules():
        for child_name, child in list(module.named_children()):
            if type(child) is nn.LayerNorm and len(child.normalized_shape) == 1:
                fused = FusedLayerNorm(child.normalized_shape, eps=child.eps, elementwise_affine=True)
                fused.weight = child.weight
                fused.bias = child.bias
                setattr(module, child_name, fused)
            elif child.__class__.__name__ == "RMSNorm" and hasattr(child, "weight") and hasattr(child, "eps"):
                fused = FusedRMSNorm(child.weight.numel(), eps=child.eps)
                fused.weight = child.weight
                setattr(module, child_name, fused)
    return model


class _FusedDropoutAddLNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, z, weight, bias, eps, p):
        ext = _load_extension(required=True)
        y, s, mask, mean, rstd = ext.dropout_add_ln_fwd(x, z, weight, bias, eps, p)
        ctx.save_for_backward(s, weight, mask, mean, rstd)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load_extension(required=True)
        s, weight, mask, mean, rstd = ctx.saved_tensors
        dx, dz, dw, db = ext.dropout_add_ln_bwd(dy, s, weight, mask, mean, rstd, ctx.p)
        return dx, dz, dw, db, None, None


class FusedDropoutAddLayerNorm(nn.LayerNorm):
    """``LayerNorm(x + dropout(z))`` in ONE kernel pair — the residual
    junction BERT executes twice per layer. The dropout mask comes from the
    torch philox state in-kernel (hipGraph-capture-safe) and is saved as u8
    for the backward, which produces dx and the mask-scaled dz together
    with dweight/dbias in a single pass. Parameter names/shapes match
    nn.LayerNorm, so checkpoints are interchangeable. Falls back to the
    composite ops off-GPU / non-bf16."""

    def __init__(self, normalized_shape, eps=1e-12, p=0.0):
        super().__init__(normalized_shape, eps=eps)
        self.p = p

    def forward(self, x, z=None):
        if z is None:  # plain LayerNorm use (no junction)
            return super().forward(x)
        p = self.p if self.training else 0.0
        # the single-kernel path measured 0.60x the composite at BERT
        # shapes (the u8 mask + saved-sum writes outweigh the saved
        # launches once torch's dropout/add/LN are graph-captured), so it
        # is opt-in until the mask is bit-packed:
        if (
            os.environ.get("ACCELERATE_AMD_FUSED_JUNCTION") == "1"
            and _use_fused(x, self.weight, self.bias)
            and x.shape == z.shape
            and z.dtype == x.dtype
            and len(self.normalized_shape) == 1
            and x.shape[-1] <= 2048
        ):
            return _FusedDropoutAddLNFn.apply(
                x.contiguous(), z.contiguous(), self.weight, self.bias, self.eps, p
            )
        h = x + torch.nn.functional.dropout(z, p=p, training=p > 0)
        return super().forward(h)
