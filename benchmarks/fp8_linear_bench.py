"""fp8 vs bf16 training throughput on GPT-2-large-class blocks
(BASELINE config #5: fp8 ≥ +25% token throughput over bf16).

Measures fwd+bwd tokens/s through a stack of transformer MLP+attention
projection GEMMs at GPT-2-large shapes (hidden 1280, ffn 5120, seq 1024)
with FP8Linear (e4m3 fwd / e5m2 dgrad via hipBLASLt) vs bf16 nn.Linear.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn


def build_stack(hidden, ffn, n_blocks, fp8):
    from accelerate_amd.ops.fp8 import convert_linears_to_fp8

    layers = []
    for _ in range(n_blocks):
        layers += [nn.Linear(hidden, ffn), nn.GELU(), nn.Linear(ffn, hidden)]
    model = nn.Sequential(*layers).cuda().to(torch.bfloat16)
    if fp8:
        from accelerate_amd.utils.dataclasses import FP8RecipeKwargs

        convert_linears_to_fp8(model, FP8RecipeKwargs(use_first_last_bf16=False))
    return model


def run(model, tokens, hidden, iters=30, warmup=10):
    """hipGraph-captured fwd+bwd so both variants measure GPU time, not
    python autograd overhead (training runs under graphs too)."""
    import gc

    x = torch.randn(tokens, hidden, device="cuda", dtype=torch.bfloat16, requires_grad=True)

    def step():
        model(x).sum().backward()

    for _ in range(warmup):
        step()
    gc.collect()
    torch.cuda.synchronize()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            step()
    torch.cuda.current_stream().wait_stream(side)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        g.replay()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return tokens / dt


def main():
    hidden, ffn, n_blocks, tokens = 1280, 5120, 12, 8192  # GPT-2-large shapes, seq 8*1024
    torch.manual_seed(0)
    bf16_tps = run(build_stack(hidden, ffn, n_blocks, fp8=False), tokens, hidden)
    torch.manual_seed(0)
    fp8_tps = run(build_stack(hidden, ffn, n_blocks, fp8=True), tokens, hidden)
    result = {
        "bench": "fp8_vs_bf16_linear_stack (GPT-2-large shapes)",
        "bf16_tokens_per_s": round(bf16_tps),
        "fp8_tokens_per_s": round(fp8_tps),
        "speedup": round(fp8_tps / bf16_tps, 3),
        "config": {"hidden": hidden, "ffn": ffn, "blocks": n_blocks, "tokens": tokens},
    }
    print(json.dumps(result))


if __name__ == "__main__":
    main()
