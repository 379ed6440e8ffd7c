#!/usr/bin/env python
"""Headline benchmark: samples/sec, nlp_example BERT-base, DDP bf16
(BASELINE.json metric) on 1..8 MI355X.

Synthetic MRPC-shaped data (seq_len 128, random tokens), random-init
BERT-base, per-GPU batch 16 (examples/nlp_example.py:43 MAX_GPU_BATCH_SIZE),
AdamW lr 2e-5 + linear-warmup schedule — the nlp_example training loop run
through accelerate_amd's Accelerator/prepare/backward with our RCCL reducer
and fused CDNA4 AdamW.

Launched by the driver as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
Rank 0 prints one JSON line.
"""

import argparse
import json
import os
import time

import torch
from torch.utils.data import DataLoader, TensorDataset

PER_GPU_BATCH = 16
SEQ_LEN = 128


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--profile", action="store_true", help="emit a chrome trace under gpurun_out/")
    parser.add_argument(
        "--bucket-mb",
        type=int,
        default=int(os.environ.get("BENCH_BUCKET_MB", "64")),
        help="DDP reducer bucket size in MiB (sweep 32/64/128 on the first 8-GPU lease)",
    )
    args = parser.parse_args()

    from accelerate_amd import Accelerator, set_seed
    from accelerate_amd.models import BertConfig, BertForSequenceClassification
    from accelerate_amd.utils.dataclasses import DistributedDataParallelKwargs

    on_gpu_env = torch.cuda.is_available()
    # hipBLASLt algorithm selection pre-tuned on MI355X (TunableOp CSV,
    # produced by benchmarks/tune_gemms.py) — read-only at runtime
    tun_csv = os.path.join(os.path.dirname(os.path.abspath(__file__)), "profiles", "tunableop_mi355x.csv")
    if on_gpu_env and os.path.exists(tun_csv) and os.environ.get("BENCH_TUNABLEOP", "1") == "1":
        try:
            import torch.cuda.tunable as tunable

            tunable.enable(True)
            tunable.tuning_enable(False)
            tunable.read_file(tun_csv)
        except Exception as e:
            print(f"[bench] TunableOp unavailable: {e}", flush=True)
    # bf16-native weights + fp32 master in the fused optimizer (default):
    # no autocast cast kernels, bf16 gradient all-reduce over xGMI
    bf16_weights = on_gpu_env and os.environ.get("BENCH_BF16_WEIGHTS", "1") == "1"
    mixed = "no" if bf16_weights or not on_gpu_env else "bf16"
    accelerator = Accelerator(
        mixed_precision=mixed,
        kwargs_handlers=[DistributedDataParallelKwargs(bucket_cap_mb=args.bucket_mb)],
    )
    set_seed(42)

    n = accelerator.num_processes
    device = accelerator.device
    on_gpu = device.type == "cuda"

    if os.environ.get("BENCH_TINY") == "1":
        # CI plumbing smoke ONLY (2-proc gloo test): tiny model, NOT the
        # measured configuration — the emitted config block says so
        config = BertConfig(num_hidden_layers=2, hidden_size=128, num_attention_heads=2,
                            intermediate_size=256)
    else:
        config = BertConfig.bert_base()
    model = BertForSequenceClassification(config)
    if bf16_weights:
        model = model.to(torch.bfloat16)

    if on_gpu:
        from accelerate_amd.ops.optim import FusedAdamW

        optimizer = FusedAdamW(model.parameters(), lr=2e-5, weight_decay=0.01)
    else:
        optimizer = torch.optim.AdamW(model.parameters(), lr=2e-5, weight_decay=0.01)

    total_steps = args.steps + args.warmup
    n_samples = (total_steps + 2) * PER_GPU_BATCH * max(n, 1)
    g = torch.Generator().manual_seed(1234)
    input_ids = torch.randint(0, config.vocab_size, (n_samples, SEQ_LEN), generator=g)
    token_type_ids = torch.zeros(n_samples, SEQ_LEN, dtype=torch.long)
    attention_mask = torch.ones(n_samples, SEQ_LEN, dtype=torch.long)
    labels = torch.randint(0, 2, (n_samples,), generator=g)
    dataset = TensorDataset(input_ids, attention_mask, token_type_ids, labels)
    dataloader = DataLoader(dataset, batch_size=PER_GPU_BATCH, shuffle=True, drop_last=True)

    scheduler = torch.optim.lr_scheduler.LambdaLR(optimizer, lambda step: min(1.0, (step + 1) / 100))

    model, optimizer, dataloader, scheduler = accelerator.prepare(model, optimizer, dataloader, scheduler)
    model.train()

    data_iter = iter(dataloader)

    def next_batch():
        nonlocal data_iter
        try:
            return next(data_iter)
        except StopIteration:
            data_iter = iter(dataloader)
            return next(data_iter)

    use_graph = on_gpu and os.environ.get("BENCH_GRAPH", "1") == "1"
    step_mode = "eager"

    def eager_step(batch):
        ids, mask, types, lbl = batch
        optimizer.zero_grad(set_to_none=False)
        out = model(ids, attention_mask=mask, token_type_ids=types, labels=lbl)
        accelerator.backward(out["loss"])
        optimizer.step()
        scheduler.step()
        return out["loss"]

    one_step = eager_step

    if use_graph:
        # Whole-step hipGraph: zero_grad + fwd + bwd (+ RCCL allreduce) +
        # fused AdamW replay from ONE graph launch. Inputs live in static
        # device buffers; lr is a device scalar the scheduler refreshes.
        graph = loss_buf = None
        try:
            static = [t.to(device, non_blocking=True) for t in next_batch()]

            def graph_body():
                # set_to_none INSIDE the capture: graph-pool allocations replay
                # at stable addresses, so backward ASSIGNS grads (no zero-fill
                # kernels, no accumulate-add kernels — ~400 launches/step saved)
                optimizer.zero_grad(set_to_none=True)
                out = model(static[0], attention_mask=static[1], token_type_ids=static[2], labels=static[3])
                accelerator.backward(out["loss"])
                optimizer.step()
                return out["loss"]

            # eager warmup on a side stream (materializes grads/opt state/buckets)
            import gc

            gc.collect()  # stale autograd graphs break ROCm capture_end
            torch.cuda.synchronize()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    graph_body()
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                loss_buf = graph_body()
            captured = True
        except Exception as e:
            print(f"[bench] hipGraph capture failed ({e}); falling back to eager", flush=True)
            captured = False

        # all-or-none: every rank must run the SAME collective sequence, so a
        # capture failure anywhere sends the whole job down the eager path
        if accelerator.use_distributed:
            oks = [None] * n
            torch.distributed.all_gather_object(oks, captured)
            captured = all(oks)

        if captured:
            # sanity replay (all ranks together): RCCL-in-graph exercised
            # before committing; finiteness agreed across ranks
            graph.replay()
            torch.cuda.synchronize()
            finite = bool(torch.isfinite(loss_buf).all().item())
            if accelerator.use_distributed:
                oks = [None] * n
                torch.distributed.all_gather_object(oks, finite)
                finite = all(oks)
            captured = finite
            if not finite:
                print("[bench] captured step produced non-finite loss; eager fallback", flush=True)

        if captured:
            fused = optimizer.optimizer  # the FusedAdamW under the wrapper

            def graph_step(batch):
                for dst, src in zip(static, batch):
                    dst.copy_(src, non_blocking=True)
                scheduler.step()
                fused.refresh_hyperparams()
                graph.replay()
                return loss_buf

            one_step = graph_step
            step_mode = "hipgraph"
        else:
            one_step = eager_step

    # warmup
    for _ in range(args.warmup):
        one_step(next_batch())

    accelerator.wait_for_everyone()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step(next_batch())
    if on_gpu:
        torch.cuda.synchronize()
    accelerator.wait_for_everyone()
    t1 = time.perf_counter()

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if accelerator.use_distributed:
        elapsed = elapsed.to(device if on_gpu else "cpu")
        torch.distributed.all_reduce(elapsed, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(elapsed.item())

    if accelerator.is_main_process:
        global_batch = PER_GPU_BATCH * n
        samples_per_sec = global_batch * args.steps / elapsed
        result = {
            "metric": "samples/sec (whole node) on nlp_example BERT-base, DDP bf16",
            "value": round(samples_per_sec, 2),
            "unit": "samples/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32(cpu-ci)",
            "data": "synthetic",
            "config": {
                "model": "bert-tiny(ci-smoke)" if os.environ.get("BENCH_TINY") == "1" else "bert-base",
                "global_batch": global_batch,
                "per_gpu_batch": PER_GPU_BATCH,
                "seq_len": SEQ_LEN,
                "parallelism": f"dp{n}",
                "optimizer": "fused_adamw_hip" if on_gpu else "torch_adamw",
                "step_mode": step_mode,
                "bucket_mb": args.bucket_mb,
                "weights": "bf16+fp32_master" if bf16_weights else ("fp32+autocast_bf16" if on_gpu else "fp32"),
            },
        }
        print(json.dumps(result))
    accelerator.end_training()


if __name__ == "__main__":
    main()
