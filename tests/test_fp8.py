"""fp8 path tests (gpu): cast+amax kernel numerics, FP8Linear vs bf16
reference, convert_linears_to_fp8 layer selection."""

import pytest
import torch
import torch.nn as nn

gpu = pytest.mark.gpu


def test_convert_selection_cpu():
    from accelerate_amd.ops.fp8 import FP8Linear, convert_linears_to_fp8

    model = nn.Sequential(nn.Linear(64, 64), nn.Linear(64, 64), nn.Linear(64, 64))
    convert_linears_to_fp8(model)
    # first/last stay bf16 Linears
    assert type(model[0]) is nn.Linear
    assert isinstance(model[1], FP8Linear)
    assert type(model[2]) is nn.Linear


@gpu
def test_fp8_cast_amax_kernel():
    from accelerate_amd.ops import _load_extension

    ext = _load_extension(required=True)
    x = torch.randn(4096 + 13, device="cuda", dtype=torch.bfloat16) * 3
    scale = torch.ones(1, device="cuda")
    amax = torch.zeros(1, device="cuda")
    out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device="cuda")
    ext.fp8_cast_amax(x, out, scale, amax, False)
    torch.cuda.synchronize()
    assert torch.allclose(amax, x.float().abs().max().reshape(1), rtol=1e-3)
    # quantized values match torch's own cast at scale 1
    ref = x.to(torch.float8_e4m3fn)
    assert torch.equal(out.view(torch.uint8), ref.view(torch.uint8))


@gpu
def test_fp8_cast_scale_applied():
    from accelerate_amd.ops import _load_extension

    ext = _load_extension(required=True)
    x = torch.full((1024,), 2.0, device="cuda", dtype=torch.bfloat16)
    scale = torch.tensor([4.0], device="cuda")
    amax = torch.zeros(1, device="cuda")
    out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device="cuda")
    ext.fp8_cast_amax(x, out, scale, amax, False)
    torch.cuda.synchronize()
    assert torch.allclose(out.float(), torch.full_like(x, 8.0).float())
    assert amax.item() == 2.0


@gpu
def test_scaled_mm_available():
    a = torch.randn(64, 128, device="cuda", dtype=torch.bfloat16).to(torch.float8_e4m3fn)
    b = torch.randn(256, 128, device="cuda", dtype=torch.bfloat16).to(torch.float8_e4m3fn)
    s = torch.ones(1, device="cuda")
    y = torch._scaled_mm(a, b.t(), scale_a=s, scale_b=s, out_dtype=torch.bfloat16)
    assert y.shape == (64, 256)


@gpu
def test_fp8_linear_close_to_bf16():
    from accelerate_amd.ops.fp8 import FP8Linear

    torch.manual_seed(0)
    lin = nn.Linear(256, 512, device="cuda", dtype=torch.bfloat16)
    fp8 = FP8Linear.from_linear(lin)
    x = torch.randn(64, 256, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    # first call establishes amax history (scale=1); second uses scaled values
    for _ in range(2):
        y8 = fp8(x)
    y16 = nn.functional.linear(x2, lin.weight, lin.bias)
    rel = (y8.float() - y16.float()).abs().mean() / y16.float().abs().mean()
    assert rel < 0.08, f"fp8 forward too far from bf16: rel={rel}"
    # backward produces gradients of the right shape and finite values
    y8.sum().backward()
    torch.cuda.synchronize()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert lin.weight.grad is not None and torch.isfinite(lin.weight.grad).all()


@gpu
def test_fp8_training_tracks_bf16():
    """fp8 training must track the bf16 loss trajectory (same model/seed),
    the reference's fp8-vs-bf16 quality contract."""
    from accelerate_amd.ops.fp8 import convert_linears_to_fp8

    def run(fp8):
        torch.manual_seed(0)
        model = (
            nn.Sequential(nn.Linear(64, 256), nn.ReLU(), nn.Linear(256, 256), nn.ReLU(), nn.Linear(256, 16))
            .cuda()
            .to(torch.bfloat16)
        )
        if fp8:
            convert_linears_to_fp8(model)
        opt = torch.optim.SGD(model.parameters(), lr=0.05)
        g = torch.Generator(device="cuda").manual_seed(7)
        x = torch.randn(128, 64, device="cuda", dtype=torch.bfloat16)
        y = torch.randn(128, 16, device="cuda", dtype=torch.bfloat16)
        losses = []
        for _ in range(30):
            opt.zero_grad()
            loss = ((model(x) - y) ** 2).float().mean()
            loss.backward()
            opt.step()
            losses.append(loss.item())
        return losses

    bf16_losses = run(False)
    fp8_losses = run(True)
    assert fp8_losses[-1] < bf16_losses[-1] * 1.05, f"fp8 diverges from bf16: {fp8_losses[-1]} vs {bf16_losses[-1]}"
    assert fp8_losses[-1] < fp8_losses[0], "fp8 loss must decrease"


@gpu
def test_fp8_cast_transpose_kernel():
    from accelerate_amd.ops import _load_extension

    ext = _load_extension(required=True)
    x = torch.randn(256, 384, device="cuda", dtype=torch.bfloat16) * 2
    scale = torch.ones(1, device="cuda")
    amax = torch.zeros(1, device="cuda")
    out = torch.empty(256, 384, dtype=torch.float8_e4m3fn, device="cuda")
    out_t = torch.empty(384, 256, dtype=torch.float8_e4m3fn, device="cuda")
    ext.fp8_cast_transpose(x, out, out_t, scale, amax, False)
    torch.cuda.synchronize()
    ref = x.to(torch.float8_e4m3fn)
    assert torch.equal(out.view(torch.uint8), ref.view(torch.uint8))
    assert torch.equal(out_t.view(torch.uint8), ref.t().contiguous().view(torch.uint8))
    assert torch.allclose(amax, x.float().abs().max().reshape(1), rtol=1e-3)


@gpu
def test_accelerator_fp8_end_to_end():
    """mixed_precision='fp8' through Accelerator.prepare: linears converted,
    loss decreases, grads finite."""
    import torch.nn as nn

    from accelerate_amd import Accelerator
    from accelerate_amd.ops.fp8 import FP8Linear

    torch.manual_seed(0)
    acc = Accelerator(mixed_precision="fp8")
    model = nn.Sequential(
        nn.Linear(128, 256), nn.GELU(), nn.Linear(256, 256), nn.GELU(), nn.Linear(256, 128)
    ).to(torch.bfloat16)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    model, opt = acc.prepare(model, opt)
    inner = acc.unwrap_model(model, keep_fp32_wrapper=False)
    assert any(isinstance(m, FP8Linear) for m in inner.modules()), "fp8 conversion did not run"
    x = torch.randn(256, 128, device="cuda", dtype=torch.bfloat16)
    # learnable target (a fixed linear map of x): the loss is clearly reducible
    w_true = torch.randn(128, 128, device="cuda", dtype=torch.bfloat16) * 0.2
    y = x @ w_true
    losses = []
    for _ in range(30):
        opt.zero_grad()
        loss = ((model(x) - y) ** 2).float().mean()
        acc.backward(loss)
        opt.step()
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses)))
    # quality is pinned by test_fp8_training_tracks_bf16; here we assert the
    # Accelerator wiring trains (loss visibly reduced on a learnable target)
    assert losses[-1] < losses[0] * 0.9, (losses[0], losses[-1])
