"""Microbenchmarks: (a) hipBLASLt fp8 scaled_mm vs bf16 matmul at training
shapes; (b) optional pure-GEMM TunableOp tuning for the bench shapes
(BENCH_TUNE=1)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def fp8_vs_bf16():
    s = torch.ones(1, device="cuda")
    for (m, k, n) in [(8192, 1280, 5120), (8192, 5120, 1280), (16384, 4096, 4096), (4096, 4096, 4096), (2048, 768, 3072)]:
        a16 = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        b16 = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        a8 = a16.to(torch.float8_e4m3fn)
        b8 = b16.to(torch.float8_e4m3fn)
        t_bf16 = timeit(lambda: a16 @ b16.t())
        t_fp8 = timeit(lambda: torch._scaled_mm(a8, b8.t(), scale_a=s, scale_b=s, out_dtype=torch.bfloat16))
        fl = 2 * m * k * n
        print(
            f"M{m} K{k} N{n}: bf16 {t_bf16*1e6:7.1f}us ({fl/t_bf16/1e12:6.0f} TF)  "
            f"fp8 {t_fp8*1e6:7.1f}us ({fl/t_fp8/1e12:6.0f} TF)  ratio {t_bf16/t_fp8:.2f}x",
            flush=True,
        )


def tune():
    import torch.cuda.tunable as tunable

    tunable.enable(True)
    tunable.tuning_enable(True)
    tunable.set_max_tuning_iterations(30)
    shapes = [
        (2048, 768, 2304), (2048, 768, 768), (2048, 768, 3072), (2048, 3072, 768),
        (2048, 2304, 768), (2048, 3072, 768),
    ]
    for (m, k, n) in shapes:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16)
        (a @ b).sum().item()
        print(f"tuned {m}x{k}x{n}", flush=True)
    out = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "profiles", "tunableop_mi355x.csv")
    tunable.write_file(out)
    print("wrote", out, flush=True)


if __name__ == "__main__":
    fp8_vs_bf16()
    if os.environ.get("BENCH_TUNE", "0") == "1":
        tune()
