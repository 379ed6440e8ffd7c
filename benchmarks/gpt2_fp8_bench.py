"""GPT-2-large fp8 vs bf16 training throughput (BASELINE config #5).

The reference's comparable headline: FSDP2 + torchao fp8 gives ~+25% token
throughput over bf16 on 8xH100 (examples/torch_native_parallelism/README).
Here: one MI355X, synthetic data, random-init GPT-2-large (774M), fused
CDNA4 AdamW, fp8 Linears = OCP e4m3/e5m2 delayed scaling over hipBLASLt
scaled GEMM with our fused cast(+transpose)+amax kernels.

  python benchmarks/gpt2_fp8_bench.py [--seq 1024] [--batch 24] [--steps 6]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run(mode: str, args) -> dict:
    from accelerate_amd import Accelerator, set_seed
    from accelerate_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel
    from accelerate_amd.ops.optim import FusedAdamW
    from accelerate_amd.state import AcceleratorState

    AcceleratorState._reset_state()
    acc = Accelerator(mixed_precision=mode)
    set_seed(0)
    cfg = GPT2Config.gpt2_large(max_position_embeddings=args.seq)
    model = GPT2LMHeadModel(cfg).to(torch.bfloat16)
    opt = FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.01)
    model, opt = acc.prepare(model, opt)

    g = torch.Generator().manual_seed(7)
    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), generator=g).to(acc.device)

    def step():
        opt.zero_grad(set_to_none=True)
        loss = model(ids, labels=ids)["loss"]
        acc.backward(loss)
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    toks = args.batch * args.seq / dt
    return {"mode": mode, "tokens_per_s": round(toks, 1), "ms_per_step": round(dt * 1e3, 1)}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--batch", type=int, default=24)
    p.add_argument("--steps", type=int, default=6)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--modes", default="no,fp8", help="comma list of: no (bf16-native), bf16, fp8")
    args = p.parse_args()

    out = {"bench": "gpt2-large train step (1x MI355X)", "seq": args.seq, "batch": args.batch}
    rows = [run(m, args) for m in args.modes.split(",")]
    out["rows"] = rows
    by = {r["mode"]: r["tokens_per_s"] for r in rows}
    if "fp8" in by and ("no" in by or "bf16" in by):
        base = by.get("no", by.get("bf16"))
        out["fp8_speedup"] = round(by["fp8"] / base, 3)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
