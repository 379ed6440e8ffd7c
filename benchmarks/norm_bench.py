"""Fused norm kernels vs torch eager at the shapes that matter:
BERT LayerNorm (d=768) and Llama RMSNorm (d=4096) fwd+bwd.

Run on a GPU box:  python benchmarks/norm_bench.py
"""

import time

import torch
import torch.nn as nn

from accelerate_amd.ops.norms import FusedLayerNorm, FusedRMSNorm


def bench(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def run(shape, mod_fused, mod_ref, tag):
    x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn_like(x)

    def step(m):
        xr = x.detach().requires_grad_(True)
        y = m(xr)
        y.backward(dy)
        return xr.grad

    fused_us = bench(lambda: step(mod_fused))
    ref_us = bench(lambda: step(mod_ref))
    # fwd-only
    with torch.no_grad():
        f_fwd = bench(lambda: mod_fused(x))
        r_fwd = bench(lambda: mod_ref(x))
    n_bytes = x.numel() * 2
    print(
        f"{tag:28s} fused {fused_us:8.1f} us  torch {ref_us:8.1f} us  "
        f"speedup {ref_us / fused_us:4.2f}x | fwd {f_fwd:6.1f}/{r_fwd:6.1f} us "
        f"| bwd eff.BW {(5 * n_bytes) / ((fused_us - f_fwd) * 1e-6) / 1e12:5.2f} TB/s"
    )


def main():
    torch.manual_seed(0)
    # BERT-large step shape: B32 x S512, d=768
    d = 768
    ln = FusedLayerNorm(d).cuda().to(torch.bfloat16)
    ref = nn.LayerNorm(d).cuda().to(torch.bfloat16)
    run((32 * 512, d), ln, ref, f"LayerNorm {32 * 512}x{d}")

    # Llama-8B shape: B4 x S2048, d=4096 (split dx/dw path)
    d = 4096
    rms = FusedRMSNorm(d).cuda().to(torch.bfloat16)

    class RefRMS(nn.Module):
        def __init__(self):
            super().__init__()
            self.weight = nn.Parameter(torch.ones(d, device="cuda", dtype=torch.bfloat16))

        def forward(self, x):
            xf = x.float()
            xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
            return self.weight * xf.to(x.dtype)

    run((4 * 2048, d), rms, RefRMS(), f"RMSNorm {4 * 2048}x{d}")
    # long-sequence shape (CP regime)
    run((2 * 8192, d), rms, RefRMS(), f"RMSNorm {2 * 8192}x{d}")
    # d=1024 KEEP path
    d = 1024
    ln2 = FusedLayerNorm(d).cuda().to(torch.bfloat16)
    ref2 = nn.LayerNorm(d).cuda().to(torch.bfloat16)
    run((32 * 512, d), ln2, ref2, f"LayerNorm {32 * 512}x{d}")


if __name__ == "__main__":
    import sys
    if "--junction" not in sys.argv:
        main()


def junction_bench():
    """Fused dropout+add+LN vs composite at the BERT bench shape."""
    from accelerate_amd.ops.norms import FusedDropoutAddLayerNorm

    rows, d, p = 2048, 768, 0.1
    fused = FusedDropoutAddLayerNorm(d, eps=1e-12, p=p).cuda().to(torch.bfloat16)
    fused.train()
    ln = nn.LayerNorm(d, eps=1e-12).cuda().to(torch.bfloat16)
    x = torch.randn(rows, d, device="cuda", dtype=torch.bfloat16)
    z = torch.randn(rows, d, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(rows, d, device="cuda", dtype=torch.bfloat16)

    def step_fused():
        xr = x.detach().requires_grad_(True)
        zr = z.detach().requires_grad_(True)
        fused(xr, zr).backward(dy)

    def step_comp():
        xr = x.detach().requires_grad_(True)
        zr = z.detach().requires_grad_(True)
        ln(xr + torch.nn.functional.dropout(zr, 0.1, True)).backward(dy)

    f = bench(step_fused)
    c = bench(step_comp)
    print(f"junction {rows}x{d} p={p}: fused {f:7.1f} us  composite {c:7.1f} us  speedup {c / f:4.2f}x")


if __name__ == "__main__" and "--junction" in __import__("sys").argv:
    junction_bench()
