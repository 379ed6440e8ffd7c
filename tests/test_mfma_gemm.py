"""Hand-written MFMA GEMM numerics vs fp32 torch reference (gpu)."""

import pytest
import torch

gpu = pytest.mark.gpu


@gpu
@pytest.mark.parametrize("shape", [(256, 64, 128), (2048, 768, 3072), (512, 512, 512), (128, 1024, 128)])
def test_mfma_gemm_bt_matches_fp32(shape):
    from accelerate_amd.ops import _load_extension

    ext = _load_extension(required=True)
    M, K, N = shape
    torch.manual_seed(1)
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda").to(torch.bfloat16)
    c = ext.mfma_gemm_bt(a, b, bias)
    ref = a.float() @ b.float().t() + bias.float()
    rel = (c.float() - ref).abs().max() / ref.abs().max()
    assert rel < 0.02, rel
    # no-bias path
    c2 = ext.mfma_gemm_bt(a, b, None)
    ref2 = a.float() @ b.float().t()
    assert (c2.float() - ref2).abs().max() / ref2.abs().max() < 0.02


@gpu
@pytest.mark.parametrize("shape", [(256, 128, 256), (512, 384, 768), (1024, 128, 256)])
def test_gemm8_256sq_8phase_matches_fp32(shape):
    """Shapes hitting the 8-phase 256^2 dispatch (M,N%256==0, K%128==0),
    with and without bias."""
    from accelerate_amd.ops import _load_extension

    ext = _load_extension(required=True)
    M, K, N = shape
    torch.manual_seed(3)
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda").to(torch.bfloat16)
    c = ext.mfma_gemm_bt(a, b, bias)
    ref = a.float() @ b.float().t() + bias.float()
    assert (c.float() - ref).abs().max() / ref.abs().max() < 0.02
    c2 = ext.mfma_gemm_bt(a, b, None)
    ref2 = a.float() @ b.float().t()
    assert (c2.float() - ref2).abs().max() / ref2.abs().max() < 0.02
