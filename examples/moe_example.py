"""Mixture-of-experts training with expert parallelism (EP).

Each rank owns n_experts/world experts; tokens are dispatched by
variable-split all-to-all (dropless). The gate/attention/backbone are
data-parallel through the DDP engine; expert params carry `_no_ddp_sync`.

Run:
  python examples/moe_example.py --cpu            # 1 process
  python -m accelerate_amd launch --num_processes 8 examples/moe_example.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models.llama_moe import LlamaMoEConfig, LlamaMoEForCausalLM


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cpu", action="store_true")
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--mixed_precision", default=None)
    args = p.parse_args()

    acc = Accelerator(cpu=args.cpu, mixed_precision=args.mixed_precision)
    set_seed(0)
    model = LlamaMoEForCausalLM(LlamaMoEConfig.tiny_moe(n_experts=max(4, 2 * acc.num_processes)))
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model, opt = acc.prepare(model, opt)

    g = torch.Generator().manual_seed(1)
    for step in range(args.steps):
        ids = torch.randint(0, 1024, (4, 64), generator=g).to(acc.device)
        opt.zero_grad()
        out = model(ids, labels=ids)
        acc.backward(out["loss"])  # aux balance loss is inside out["loss"]
        opt.step()
        if step % 5 == 0:
            acc.print(f"step {step}: loss {out['loss'].item():.3f} aux {out['aux_loss'].item():.4f}")
    acc.end_training()


if __name__ == "__main__":
    main()
