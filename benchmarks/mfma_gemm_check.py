import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from accelerate_amd.ops import _load_extension
ext = _load_extension(required=True)
torch.manual_seed(0)
# refcheck at several shapes
for (M, K, N) in [(256, 64, 128), (256, 256, 256), (2048, 768, 3072), (2048, 768, 768)]:
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda").to(torch.bfloat16)
    c = ext.mfma_gemm_bt(a, b, bias)
    ref = (a.float() @ b.float().t() + bias.float()).to(torch.bfloat16)
    d = (c.float() - ref.float()).abs().max().item()
    rel = d / ref.float().abs().max().item()
    print(f"M{M} K{K} N{N}: maxdiff {d:.4f} rel {rel:.5f} {'OK' if rel < 0.02 else 'FAIL'}")
# perf at bench shapes + 4096³
def t(f, iters=50):
    for _ in range(10): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters
for (M, K, N) in [(2048, 768, 3072), (2048, 768, 2304), (2048, 3072, 768), (4096, 4096, 4096)]:
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    d_ours = t(lambda: ext.mfma_gemm_bt(a, b, None))
    d_blas = t(lambda: a @ b.t())
    fl = 2*M*K*N
    print(f"M{M} K{K} N{N}: ours {d_ours*1e6:7.1f}us ({fl/d_ours/1e12:5.0f} TF) | hipBLASLt {d_blas*1e6:7.1f}us ({fl/d_blas/1e12:5.0f} TF)")
