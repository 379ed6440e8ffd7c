"""Fused LayerNorm/RMSNorm numerics vs plain fp32 torch reference (gpu)."""

import pytest
import torch
import torch.nn as nn

gpu = pytest.mark.gpu


def test_fused_norm_cpu_fallback():
    from accelerate_amd.ops.norms import FusedLayerNorm, FusedRMSNorm

    ln = FusedLayerNorm(64)
    x = torch.randn(4, 64)
    ref = nn.LayerNorm(64)
    assert torch.allclose(ln(x), ref(x), atol=1e-6)
    rms = FusedRMSNorm(64)
    y = rms(x)
    assert y.shape == x.shape


@gpu
@pytest.mark.parametrize("shape", [(2048, 768), (16, 128, 768), (64, 4096)])
def test_fused_layernorm_matches_fp32(shape):
    from accelerate_amd.ops.norms import FusedLayerNorm

    torch.manual_seed(0)
    d = shape[-1]
    x32 = torch.randn(*shape, device="cuda", requires_grad=True)
    x16 = x32.detach().to(torch.bfloat16).requires_grad_(True)
    ref = nn.LayerNorm(d, eps=1e-5).cuda()
    fused = FusedLayerNorm(d, eps=1e-5).cuda().to(torch.bfloat16)
    with torch.no_grad():
        w = torch.randn(d, device="cuda") * 0.1 + 1.0
        b = torch.randn(d, device="cuda") * 0.1
        ref.weight.copy_(w), ref.bias.copy_(b)
        fused.weight.copy_(w.to(torch.bfloat16)), fused.bias.copy_(b.to(torch.bfloat16))

    y_ref = ref(x32)
    y = fused(x16)
    assert (y.float() - y_ref).abs().max() < 0.05, (y.float() - y_ref).abs().max()

    dy = torch.randn_like(y_ref)
    y_ref.backward(dy)
    y.backward(dy.to(torch.bfloat16))
    torch.cuda.synchronize()
    assert (x16.grad.float() - x32.grad).abs().max() < 0.05
    rel_w = (fused.weight.grad.float() - ref.weight.grad).abs().max() / ref.weight.grad.abs().max()
    rel_b = (fused.bias.grad.float() - ref.bias.grad).abs().max() / (ref.bias.grad.abs().max() + 1e-6)
    assert rel_w < 0.05, rel_w
    assert rel_b < 0.05, rel_b


@gpu
@pytest.mark.parametrize("shape", [(2048, 768), (8, 256, 4096)])
def test_fused_rmsnorm_matches_fp32(shape):
    from accelerate_amd.ops.norms import FusedRMSNorm

    torch.manual_seed(0)
    d = shape[-1]
    x32 = torch.randn(*shape, device="cuda", requires_grad=True)
    x16 = x32.detach().to(torch.bfloat16).requires_grad_(True)
    w = (torch.randn(d, device="cuda") * 0.1 + 1.0).requires_grad_(True)
    fused = FusedRMSNorm(d).cuda().to(torch.bfloat16)
    with torch.no_grad():
        fused.weight.copy_(w.to(torch.bfloat16))

    def ref_fn(x, w):
        xf = x.float()
        return (w * (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5))).float()

    y_ref = ref_fn(x32, w)
    y = fused(x16)
    assert (y.float() - y_ref).abs().max() < 0.05

    dy = torch.randn_like(y_ref)
    y_ref.backward(dy)
    y.backward(dy.to(torch.bfloat16))
    torch.cuda.synchronize()
    assert (x16.grad.float() - x32.grad).abs().max() < 0.05
    rel_w = (fused.weight.grad.float() - w.grad).abs().max() / w.grad.abs().max()
    assert rel_w < 0.05, rel_w


@gpu
def test_convert_to_fused_norms_bert():
    from accelerate_amd.models import BertConfig, BertForSequenceClassification
    from accelerate_amd.ops.norms import FusedLayerNorm, convert_to_fused_norms

    model = BertForSequenceClassification(BertConfig(num_hidden_layers=2)).cuda().to(torch.bfloat16)
    convert_to_fused_norms(model)
    assert isinstance(model.bert.layers[0].attn_norm, FusedLayerNorm)
    ids = torch.randint(0, 30522, (2, 32), device="cuda")
    labels = torch.randint(0, 2, (2,), device="cuda")
    out = model(ids, labels=labels)
    out["loss"].backward()
    torch.cuda.synchronize()
    assert torch.isfinite(out["loss"])
