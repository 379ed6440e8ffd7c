"""Attention-kernel microbench: fused fwd+bwd on Llama-8B shapes.
Used for timing and as the rocprofv3 --pmc target (small, attention-only)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import accelerate_amd.ops.attention as fa

B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
torch.manual_seed(0)
q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
iters = int(os.environ.get("FA_ITERS", "8"))
for _ in range(2):
    out = fa.flash_attention(q, k, v, causal=True)
    out.backward(torch.randn_like(out))
    q.grad = k.grad = v.grad = None
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(iters):
    out = fa.flash_attention(q, k, v, causal=True)
    out.backward(torch.randn_like(out))
    q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
flops_fwd = 2 * B * Hq * S * S // 2 * D * 2
flops_bwd = flops_fwd // 2 * 7  # dkdv 4 + dq 3 GEMM-equivalents
print(f"fwd+bwd {dt*1e3:.2f} ms/iter, ~{(flops_fwd+flops_bwd)/dt/1e12:.0f} TF/s aggregate")
