"""Llama-3-8B training step through the sharded engine (BASELINE config #3).

On one MI355X the FULL 8B model trains without sharding tricks thanks to
288 GB HBM3E (bf16 params 16G + fp32 master 32G + Adam m/v 64G + bf16 grads
16G ≈ 128G + activations); on N GPUs the driver's torchrun launch shards it
1/N. Reports tokens/s and peak memory.

  python -m torch.distributed.run --nproc-per-node N benchmarks/fsdp_llama_bench.py --seq 4096
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--layers", type=int, default=32)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--no-ac", dest="ac", action="store_false", help="disable activation checkpointing")
    p.add_argument("--fp8", action="store_true", help="fp8 Linears (OCP e4m3/e5m2, delayed scaling)")
    p.set_defaults(ac=True)
    args = p.parse_args()

    os.environ.setdefault("ACCELERATE_USE_FSDP", "1")
    from accelerate_amd import Accelerator, set_seed
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
    from accelerate_amd.ops.optim import FusedAdamW
    from accelerate_amd.utils.dataclasses import FullyShardedDataParallelPlugin

    plugin = FullyShardedDataParallelPlugin(
        transformer_cls_names_to_wrap=["LlamaDecoderLayer"],
        activation_checkpointing=args.ac,
    )
    acc = Accelerator(mixed_precision="bf16", fsdp_plugin=plugin)
    set_seed(0)
    config = LlamaConfig.llama3_8b(num_hidden_layers=args.layers, max_position_embeddings=args.seq)
    model = LlamaForCausalLM(config)
    if args.fp8:
        from accelerate_amd.ops.fp8 import convert_linears_to_fp8

        model = convert_linears_to_fp8(model.to(torch.bfloat16))
    opt = FusedAdamW(model.parameters(), lr=1e-4)
    model, opt = acc.prepare(model, opt)

    ids = torch.randint(0, config.vocab_size, (args.batch, args.seq), device=acc.device)

    def step():
        opt.zero_grad()
        out = model(ids, labels=ids)
        acc.backward(out["loss"])
        opt.step()
        return out["loss"]

    for _ in range(args.warmup):
        step()
    acc.wait_for_everyone()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    acc.wait_for_everyone()
    dt = (time.perf_counter() - t0) / args.steps

    if acc.is_main_process:
        tokens = args.batch * args.seq * acc.num_processes
        print(
            json.dumps(
                {
                    "bench": "llama3-8b sharded-engine train step",
                    "tokens_per_s": round(tokens / dt, 1),
                    "ms_per_step": round(dt * 1000, 1),
                    "n_gpus": acc.num_processes,
                    "peak_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 1),
                    "config": {
                        "layers": args.layers,
                        "seq": args.seq,
                        "batch_per_gpu": args.batch,
                        "activation_checkpointing": args.ac,
                        "dtype": ("fp8 linears + " if args.fp8 else "") + "bf16 compute + fp32 master shards",
                    },
                }
            )
        )
    acc.end_training()


if __name__ == "__main__":
    main()
