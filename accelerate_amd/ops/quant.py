"""Weight-only quantization (int8 per-channel, int4 group-wise) for inference.

MI355X-native replacement for the reference's bitsandbytes integration
(reference utils/bnb.py:280 ``replace_with_bnb_layers`` — bitsandbytes is
CUDA-only upstream). Quantization is symmetric:

  int8: scale[o] = absmax(W[o, :]) / 127,      q in [-127, 127]
  int4: scale[o, g] = absmax(W[o, g*G:(g+1)*G]) / 7, q in [-8, 7],
        stored offset-binary (q + 8), two nibbles per byte, element 2k in
        the LOW nibble of byte k.

On a GPU the dequant runs through the gfx950 kernels in
``ops/csrc/quant_kernels.hip`` (16 elems/lane streaming) and decode-shaped
int8 matvecs (<= 8 tokens) run the fused w8a16 GEMV — the weight bytes are
read once, never materialized as bf16 in HBM. On CPU a plain torch dequant
serves unit tests.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _load_extension


def quantize_int8(w: torch.Tensor):
    """[out, in] float/bf16 -> (int8 [out, in], fp32 scale [out])."""
    w32 = w.detach().float()
    scale = w32.abs().amax(dim=1).clamp_min(1e-12) / 127.0
    q = torch.round(w32 / scale[:, None]).clamp_(-127, 127).to(torch.int8)
    return q.contiguous(), scale.contiguous()


def dequantize_int8(q: torch.Tensor, scale: torch.Tensor, dtype=torch.bfloat16):
    if q.is_cuda and dtype == torch.bfloat16 and q.size(1) % 16 == 0:
        return _load_extension(required=True).int8_dequant(q, scale)
    return (q.float() * scale[:, None]).to(dtype)


def quantize_int4(w: torch.Tensor, group_size: int = 128):
    """[out, in] -> (uint8 [out, in/2] packed nibbles, fp32 scale [out, in/G])."""
    out_f, in_f = w.shape
    if in_f % 2 != 0 or in_f % group_size != 0:
        raise ValueError(f"int4 needs in_features even and % group_size ({group_size}) == 0, got {in_f}")
    w32 = w.detach().float().reshape(out_f, in_f // group_size, group_size)
    scale = w32.abs().amax(dim=2).clamp_min(1e-12) / 7.0
    q = torch.round(w32 / scale[:, :, None]).clamp_(-8, 7).to(torch.int8) + 8
    q = q.reshape(out_f, in_f).to(torch.uint8)
    packed = (q[:, 0::2] | (q[:, 1::2] << 4)).contiguous()  # element 2k -> low nibble
    return packed, scale.reshape(out_f, -1).contiguous()


def dequantize_int4(packed: torch.Tensor, scale: torch.Tensor, group_size: int = 128,
                    dtype=torch.bfloat16):
    cols = packed.size(1) * 2
    if packed.is_cuda and dtype == torch.bfloat16 and cols % 16 == 0 and group_size % 16 == 0:
        return _load_extension(required=True).int4_dequant(packed, scale, cols, group_size)
    lo = (packed & 0xF).to(torch.int16) - 8
    hi = (packed >> 4).to(torch.int16) - 8
    q = torch.stack([lo, hi], dim=2).reshape(packed.size(0), cols).float()
    s = scale[:, :, None].expand(-1, -1, group_size).reshape(packed.size(0), cols)
    return (q * s).to(dtype)


class QuantLinear(nn.Module):
    """Linear with int8/int4 weight storage (4x/8x smaller than fp32).

    Forward dequantizes to ``compute_dtype`` and runs the library GEMM —
    except decode-shaped int8 calls on GPU, which take the fused GEMV.
    Inference-only: the quantized weight has no gradient (matches the
    reference's bnb Linear8bitLt usage under big-model inference).
    """

    def __init__(self, in_features, out_features, bias=True, bits=8, group_size=128,
                 compute_dtype=torch.bfloat16, device=None):
        super().__init__()
        if bits not in (4, 8):
            raise ValueError("bits must be 4 or 8")
        self.in_features, self.out_features = in_features, out_features
        self.bits, self.group_size = bits, group_size
        self.compute_dtype = compute_dtype
        if bits == 8:
            self.register_buffer("qweight", torch.zeros(out_features, in_features, dtype=torch.int8, device=device))
            self.register_buffer("scales", torch.ones(out_features, dtype=torch.float32, device=device))
        else:
            self.register_buffer("qweight", torch.zeros(out_features, in_features // 2, dtype=torch.uint8, device=device))
            self.register_buffer("scales", torch.ones(out_features, in_features // group_size, dtype=torch.float32, device=device))
        if bias:
            self.register_buffer("bias", torch.zeros(out_features, dtype=compute_dtype, device=device))
        else:
            self.bias = None

    @classmethod
    def from_linear(cls, linear: nn.Linear, bits=8, group_size=128, compute_dtype=torch.bfloat16):
        m = cls(linear.in_features, linear.out_features, bias=linear.bias is not None,
                bits=bits, group_size=group_size, compute_dtype=compute_dtype,
                device=linear.weight.device)
        if bits == 8:
            q, s = quantize_int8(linear.weight)
        else:
            q, s = quantize_int4(linear.weight, group_size)
        m.qweight.copy_(q)
        m.scales.copy_(s)
        if linear.bias is not None:
            m.bias.copy_(linear.bias.detach().to(compute_dtype))
        return m

    def dequantize(self) -> torch.Tensor:
        if self.bits == 8:
            return dequantize_int8(self.qweight, self.scales, self.compute_dtype)
        return dequantize_int4(self.qweight, self.scales, self.group_size, self.compute_dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.to(self.compute_dtype)
        n_tokens = x.numel() // x.shape[-1]
        if (
            self.bits == 8
            and x.is_cuda
            and n_tokens <= 8
            and self.in_features % 1024 == 0
            and self.compute_dtype == torch.bfloat16
        ):
            return _load_extension(required=True).w8a16_gemv(
                self.qweight, self.scales, x.contiguous(), self.bias
            )
        return F.linear(x, self.dequantize(), self.bias)

    def extra_repr(self):
        g = f", group_size={self.group_size}" if self.bits == 4 else ""
        return f"in_features={self.in_features}, out_features={self.out_features}, bits={self.bits}{g}"
