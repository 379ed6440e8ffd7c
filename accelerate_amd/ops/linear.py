"""Decode-path Linear: route skinny (M <= 8 rows) bf16 matmuls through the
fused HBM-bound GEMV kernel (`ext.gemv_bf16`, V_DOT2_F32_BF16 inner loop).

Single-token decode on big models is weight-bandwidth-bound; rocBLAS's
skinny-GEMM path measured only ~3.4 TB/s of the ~8 TB/s HBM3E on the 70B
demo (BENCHMARKS.md). Prefill and training shapes fall through to the
stock F.linear path untouched.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _load_extension


def fast_linear(x: torch.Tensor, weight: torch.Tensor, bias=None) -> torch.Tensor:
    """F.linear with the fused GEMV on decode shapes."""
    if (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and weight.dtype == torch.bfloat16
        and not (torch.is_grad_enabled() and (x.requires_grad or weight.requires_grad))
        and weight.is_contiguous()
        and x.numel() // x.shape[-1] <= 8
        and weight.shape[1] % 8 == 0
        and (bias is None or bias.dtype == torch.bfloat16)
    ):
        ext = _load_extension(required=True)
        return ext.gemv_bf16(x, weight, bias)
    return F.linear(x, weight, bias)


class FastLinear(nn.Linear):
    """nn.Linear with the decode GEMV on the single-token path."""

    def forward(self, x):
        return fast_linear(x, self.weight, self.bias)


def convert_linears_for_inference(model: nn.Module) -> nn.Module:
    """Swap every plain nn.Linear for FastLinear (in place, shares params)."""
    for module in model.modules():
        for name, child in list(module.named_children()):
            if type(child) is nn.Linear:
                fast = FastLinear(
                    child.in_features, child.out_features,
                    bias=child.bias is not None, device="meta",
                )
                fast.weight = child.weight
                if child.bias is not None:
                    fast.bias = child.bias
                setattr(module, name, fast)
    return model
