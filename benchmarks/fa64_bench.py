import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
import accelerate_amd.ops.attention as fa
B, Hq, Hkv, S, D = 8, 16, 16, 2048, 64
torch.manual_seed(0)
q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
for _ in range(3):
    out = fa.flash_attention(q, k, v, causal=True)
    out.backward(torch.randn_like(out)); q.grad = k.grad = v.grad = None
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(10):
    out = fa.flash_attention(q, k, v, causal=True)
    out.backward(torch.randn_like(out)); q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 10
fl = 2*B*Hq*S*S//2*D*2 * 4.5
print(f"D=64 fwd+bwd {dt*1e3:.2f} ms, ~{fl/dt/1e12:.0f} TF/s aggregate")
