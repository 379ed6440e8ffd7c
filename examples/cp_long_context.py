"""Context-parallel training example: split a long sequence across GPUs.

Two rotate strategies, both routed through plain ``prepare()``:
- ``cp_impl="allgather"``: each rank all-gathers K/V per attention (fast
  until the gathered KV stops fitting).
- ``cp_impl="ring"``: P2P KV rotation over xGMI — KV memory stays
  sequence-local, so max context scales ~cp_size x further.

Run (N ranks, sequence split N ways):
  torchrun --nproc-per-node N examples/cp_long_context.py
"""

import torch
import torch.nn.functional as F

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM


def main():
    import os

    world = int(os.environ.get("WORLD_SIZE", "1"))
    acc = Accelerator(parallelism_config=ParallelismConfig(cp_size=world, cp_impl="ring"))
    set_seed(0)
    seq_len = 256 * acc.num_processes  # total context, split cp-ways per step

    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-4)
    model, optimizer = acc.prepare(model, optimizer)

    for step in range(5):
        ids = torch.randint(0, 1024, (1, seq_len), device=acc.device)
        labels = torch.randint(0, 1024, (1, seq_len), device=acc.device)
        buffers = [ids, labels]
        optimizer.zero_grad()
        # shards both buffers along the sequence dim for this step and
        # activates the registered rotate strategy inside every attention
        with acc.maybe_context_parallel(buffers=buffers, buffer_seq_dims=[1, 1]):
            ids_l, labels_l = buffers
            logits = model(ids_l)["logits"]
            loss = F.cross_entropy(logits.reshape(-1, 1024).float(), labels_l.reshape(-1))
            acc.backward(loss)
        optimizer.step()
        acc.print(f"step {step}: loss {loss.item():.4f} (local shard {ids_l.shape[1]} of {seq_len} tokens)")

    acc.end_training()


if __name__ == "__main__":
    main()
