"""Sharded (FSDP-equivalent) Llama training example.

  # one MI355X (the full 8B fits in 288 GB HBM3E):
  python examples/fsdp_llama_example.py --layers 8
  # all 8 GPUs, 1/8 shard per GPU:
  python -m accelerate_amd launch --num_processes 8 --use_fsdp examples/fsdp_llama_example.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import os

import torch
from torch.utils.data import DataLoader, TensorDataset

os.environ.setdefault("ACCELERATE_USE_FSDP", "1")

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
from accelerate_amd.utils.dataclasses import FullyShardedDataParallelPlugin


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--layers", type=int, default=8)
    parser.add_argument("--seq", type=int, default=2048)
    parser.add_argument("--steps", type=int, default=10)
    args = parser.parse_args()

    plugin = FullyShardedDataParallelPlugin(
        transformer_cls_names_to_wrap=["LlamaDecoderLayer"],
        activation_checkpointing=True,
    )
    accelerator = Accelerator(mixed_precision="bf16", fsdp_plugin=plugin)
    set_seed(0)

    config = LlamaConfig.llama3_8b(num_hidden_layers=args.layers, max_position_embeddings=args.seq)
    model = LlamaForCausalLM(config)
    if accelerator.device.type == "cuda":
        from accelerate_amd.ops.optim import FusedAdamW

        optimizer = FusedAdamW(model.parameters(), lr=1e-4)
    else:
        optimizer = torch.optim.AdamW(model.parameters(), lr=1e-4)

    g = torch.Generator().manual_seed(1)
    ids = torch.randint(0, config.vocab_size, (64, args.seq), generator=g)
    dataloader = DataLoader(TensorDataset(ids), batch_size=1, shuffle=True)

    model, optimizer, dataloader = accelerator.prepare(model, optimizer, dataloader)

    model.train()
    for step, (batch,) in enumerate(dataloader):
        if step >= args.steps:
            break
        optimizer.zero_grad()
        out = model(batch, labels=batch)
        accelerator.backward(out["loss"])
        accelerator.clip_grad_norm_(model.parameters(), 1.0)
        optimizer.step()
        if step % 2 == 0:
            accelerator.print(f"step {step}: loss {out['loss'].item():.4f}")
    accelerator.end_training()


if __name__ == "__main__":
    main()
