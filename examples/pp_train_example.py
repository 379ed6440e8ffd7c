"""Pipeline-parallel TRAINING example (GPipe / 1F1B over P2P).

The reference can only do pipeline inference; this trains. Each rank holds
one stage of the model; activations and gradients travel point-to-point
(xGMI-adjacent under RCCL).

Run:
  python -m accelerate_amd launch --num_processes 2 examples/pp_train_example.py --cpu
  python -m accelerate_amd launch --num_processes 8 examples/pp_train_example.py --schedule 1f1b
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.parallel.pp import PipelineParallelEngine


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cpu", action="store_true")
    p.add_argument("--schedule", default="gpipe", choices=["gpipe", "1f1b"])
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--microbatches", type=int, default=8)
    args = p.parse_args()

    acc = Accelerator(cpu=args.cpu)
    set_seed(0)
    model = nn.Sequential(
        nn.Linear(64, 256), nn.GELU(),
        nn.Linear(256, 256), nn.GELU(),
        nn.Linear(256, 256), nn.GELU(),
        nn.Linear(256, 10),
    )
    engine = PipelineParallelEngine(
        model=model, num_microbatches=args.microbatches, schedule=args.schedule
    )
    opt = torch.optim.AdamW(engine.parameters(), lr=1e-3)
    loss_fn = nn.CrossEntropyLoss()

    g = torch.Generator().manual_seed(1)
    for step in range(args.steps):
        X = torch.randn(32, 64, generator=g)
        T = torch.randint(0, 10, (32,), generator=g)
        opt.zero_grad()
        loss = engine.train_step(
            inputs=X if engine.is_first else None,
            targets=T if engine.is_last else None,
            loss_fn=loss_fn if engine.is_last else None,
        )
        opt.step()
        if engine.is_last and step % 5 == 0:
            print(f"step {step}: loss {loss.item():.3f}", flush=True)
    acc.end_training()


if __name__ == "__main__":
    main()
