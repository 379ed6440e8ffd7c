"""Raw decode-GEMV bandwidth at Llama-70B shapes."""
import os, sys, time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from accelerate_amd.ops import _load_extension

ext = _load_extension(required=True)
for N, K in [(8192, 8192), (28672, 8192), (8192, 28672), (128256, 8192), (10240, 8192)]:
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(1, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        ext.gemv_bf16(x, w, None)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        ext.gemv_bf16(x, w, None)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 50
    tb = N * K * 2 / dt / 1e12
    # rocBLAS comparison
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(50):
        torch.nn.functional.linear(x, w)
    torch.cuda.synchronize()
    dtr = (time.perf_counter() - t0) / 50
    print(f"N={N:6d} K={K:6d}: fused {dt*1e6:7.1f} us ({tb:5.2f} TB/s)  rocBLAS {dtr*1e6:7.1f} us ({N*K*2/dtr/1e12:5.2f} TB/s)")
