import sys, os, torch, torch.nn as nn
sys.path.insert(0, "/root/repo")
from benchmarks.fp8_linear_bench import build_stack
m = build_stack(1280, 5120, 2, fp8=True)  # 2 blocks only
x = torch.randn(8192, 1280, device="cuda", dtype=torch.bfloat16, requires_grad=True)
for _ in range(10):
    m(x).sum().backward()
torch.cuda.synchronize()
for _ in range(5):
    m(x).sum().backward()
torch.cuda.synchronize()
