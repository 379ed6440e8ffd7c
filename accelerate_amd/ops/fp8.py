"""CDNA4 fp8 training path (OCP e4m3fn fwd / e5m2 grads) — replaces the
reference's TE/torchao/MS-AMP triple backend (reference: SURVEY.md §2.4,
§2.9 N10).

`FP8Linear` keeps the master weight in bf16/fp32 and runs the forward and
grad-input GEMMs in fp8 through hipBLASLt (``torch._scaled_mm`` — a plain
library GEMM per the MI355X rules; ~2× the bf16 MFMA rate). Casting is our
fused HIP kernel (one HBM sweep: quantize with the PREVIOUS step's scale,
record the current amax), with a TE-style per-tensor delayed-scaling recipe
(amax history + margin). grad-weight stays bf16 (avoids the fp8 transpose
round-trip; the wgrad GEMM is typically the least precision-tolerant).

First/last linears stay bf16 (reference pattern: utils/ao.py:32-92).
"""

from typing import Optional

import torch
import torch.nn as nn

from ..utils.dataclasses import FP8RecipeKwargs
from . import _load_extension

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


class _ScalingState:
    """Per-tensor delayed-scaling state (scale, inverse, amax history).

    The scale refresh is amortized over ``update_interval`` steps (TE's
    amax-update-interval pattern): between refreshes the cast kernel keeps
    atomicMax-ing into the current history slot, so the recorded amax is the
    interval max — conservative, and the hot loop stays at exactly ONE
    kernel per cast."""

    def __init__(self, device, hist_len=16, fp8_max=E4M3_MAX, margin=0, update_interval=16):
        self.scale = torch.ones(1, device=device)
        self.scale_inv = torch.ones(1, device=device)
        self.amax_history = torch.zeros(hist_len, device=device)
        self.fp8_max = fp8_max
        self.margin_pow2 = float(2**margin)
        self._slot = 0
        self._calls = 0
        self.update_interval = update_interval

    def roll_and_update(self, ext):
        self._calls += 1
        # call 2 = first refresh with real history (grads may be far from
        # scale=1), then amortized every `update_interval` calls
        if self._calls != 2 and self._calls % self.update_interval:
            return
        # amax_history[slot] was filled by the cast kernel; refresh scale
        ext.fp8_update_scale(self.amax_history, self.fp8_max, self.margin_pow2, self.scale, self.scale_inv)
        self._slot = (self._slot + 1) % self.amax_history.numel()
        self.amax_history[self._slot].zero_()

    def amax_slot(self):
        return self.amax_history[self._slot : self._slot + 1]


def _cast_fp8(x: torch.Tensor, state: _ScalingState, e5m2: bool):
    ext = _load_extension(required=True)
    out = torch.empty(x.shape, dtype=torch.float8_e5m2 if e5m2 else torch.float8_e4m3fn, device=x.device)
    ext.fp8_cast_amax(x.contiguous(), out, state.scale, state.amax_slot(), e5m2)
    return out


def _cast_transpose_fp8(x2d: torch.Tensor, state: _ScalingState, e5m2: bool):
    """One HBM pass -> (fp8 [R,C], fp8 [C,R]) via the LDS-tiled kernel;
    falls back to two casts when dims aren't 64-aligned."""
    ext = _load_extension(required=True)
    dtype = torch.float8_e5m2 if e5m2 else torch.float8_e4m3fn
    R, C = x2d.shape
    if R % 128 == 0 and C % 128 == 0:
        out = torch.empty(R, C, dtype=dtype, device=x2d.device)
        out_t = torch.empty(C, R, dtype=dtype, device=x2d.device)
        ext.fp8_cast_transpose(x2d.contiguous(), out, out_t, state.scale, state.amax_slot(), e5m2)
        return out, out_t
    out = _cast_fp8(x2d, state, e5m2)
    out_t = _cast_fp8(x2d.t().contiguous(), state, e5m2)
    return out, out_t


class _FP8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, sx, sw, sg, ext):
        in_shape = x.shape
        x2d = x.reshape(-1, in_shape[-1])
        # refresh scales BEFORE casting: the device scale scalar must stay
        # untouched between a cast and its GEMM consumers (incl. backward)
        sx.roll_and_update(ext)
        # each operand cast ONCE producing both layouts: x8 (fwd) + x8t (wgrad),
        # w8 (fwd) + w8t (dgrad)
        x8, x8t = _cast_transpose_fp8(x2d, sx, e5m2=False)
        # the WEIGHT changes once per optimizer step, not once per forward:
        # cache its fp8 casts keyed on torch's in-place version counter, so
        # microbatched/accumulated steps pay the weight cast exactly once
        cache = getattr(sw, "_w8_cache", None)
        if cache is not None and cache[0] == weight._version:
            w8, w8t = cache[1], cache[2]
        else:
            sw.roll_and_update(ext)
            w8, w8t = _cast_transpose_fp8(weight, sw, e5m2=False)
            sw._w8_cache = (weight._version, w8, w8t)
        # y = (x8 @ w8^T) * (1/sx) * (1/sw)  — hipBLASLt fp8 MFMA GEMM
        y = torch._scaled_mm(
            x8, w8.t(), scale_a=sx.scale_inv, scale_b=sw.scale_inv, bias=bias, out_dtype=torch.bfloat16
        )
        ctx.save_for_backward(w8t, x8t)
        ctx.scales = (sx, sw, sg)
        ctx.ext = ext
        ctx.has_bias = bias is not None
        return y.view(*in_shape[:-1], -1)

    @staticmethod
    def backward(ctx, grad_out):
        w8t, x8t = ctx.saved_tensors
        sx, sw, sg = ctx.scales
        ext = ctx.ext
        g2d = grad_out.reshape(-1, grad_out.shape[-1])
        sg.roll_and_update(ext)
        g8, g8t = _cast_transpose_fp8(g2d.contiguous(), sg, e5m2=True)
        # dgrad: gx[M,K] = g8[M,N] @ W[N,K]; B column-major = w8t.t()
        gx = torch._scaled_mm(
            g8, w8t.t(), scale_a=sg.scale_inv, scale_b=sw.scale_inv, out_dtype=torch.bfloat16
        )
        # wgrad (fp8): gw[N,K] = g^T[N,M] @ X[M,K]; B column-major = x8t.t()
        gw = torch._scaled_mm(
            g8t, x8t.t(), scale_a=sg.scale_inv, scale_b=sx.scale_inv, out_dtype=torch.bfloat16
        )
        gb = g2d.sum(0) if ctx.has_bias else None
        return gx.view(grad_out.shape[:-1] + (gx.shape[-1],)), gw, gb, None, None, None, None


class FP8Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True, device=None, dtype=torch.bfloat16, hist_len=16, margin=0):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features, device=device, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, device=device, dtype=dtype)) if bias else None
        nn.init.kaiming_uniform_(self.weight, a=5**0.5)
        self._sx = self._sw = self._sg = None
        self._hist_len = hist_len
        self._margin = margin

    @classmethod
    def from_linear(cls, linear: nn.Linear, hist_len=16, margin=0):
        mod = cls.__new__(cls)
        nn.Module.__init__(mod)
        mod.in_features = linear.in_features
        mod.out_features = linear.out_features
        mod.weight = linear.weight
        mod.bias = linear.bias
        mod._sx = mod._sw = mod._sg = None
        mod._hist_len = hist_len
        mod._margin = margin
        return mod

    def _lazy_states(self, device):
        if self._sx is None:
            self._sx = _ScalingState(device, self._hist_len, E4M3_MAX, self._margin)
            self._sw = _ScalingState(device, self._hist_len, E4M3_MAX, self._margin)
            self._sg = _ScalingState(device, self._hist_len, E5M2_MAX, self._margin)

    def forward(self, x):
        if not x.is_cuda or x.shape[-1] % 16 != 0 or self.out_features % 16 != 0:
            # hipBLASLt fp8 requires 16-aligned shapes; fall back to bf16 GEMM
            return nn.functional.linear(x, self.weight, self.bias)
        self._lazy_states(x.device)
        ext = _load_extension(required=True)
        x = x.to(torch.bfloat16)
        w = self.weight.to(torch.bfloat16) if self.weight.dtype != torch.bfloat16 else self.weight
        bias = self.bias.to(torch.bfloat16) if self.bias is not None else None
        return _FP8LinearFn.apply(x, w, bias, self._sx, self._sw, self._sg, ext)

    def extra_repr(self):
        return f"in_features={self.in_features}, out_features={self.out_features}, fp8=e4m3/e5m2"


def convert_linears_to_fp8(model: nn.Module, recipe: Optional[FP8RecipeKwargs] = None) -> nn.Module:
    """Swap nn.Linear → FP8Linear, keeping first/last in bf16
    (reference pattern: utils/ao.py:104/:32-92)."""
    recipe = recipe or FP8RecipeKwargs()
    linear_names = [name for name, m in model.named_modules() if type(m) is nn.Linear]
    skip = set()
    if recipe.use_first_last_bf16 and linear_names:
        skip.add(linear_names[0])
        skip.add(linear_names[-1])
    for name, module in model.named_modules():
        for child_name, child in list(module.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if type(child) is nn.Linear and full not in skip:
                if child.in_features % 16 == 0 and child.out_features % 16 == 0:
                    setattr(module, child_name, FP8Linear.from_linear(child, recipe.amax_history_len, recipe.margin))
    return model
