"""The canonical training-loop example (the reference's examples/nlp_example.py
shape, offline: synthetic MRPC-like data instead of Hub downloads).

Run:
  python examples/nlp_example.py                              # 1 process
  python -m accelerate_amd launch --num_processes 8 examples/nlp_example.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import torch
from torch.utils.data import DataLoader, TensorDataset

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models import BertConfig, BertForSequenceClassification

MAX_GPU_BATCH_SIZE = 16
EVAL_BATCH_SIZE = 32


def get_dataloaders(batch_size: int, seq_len: int = 128, n_train: int = 1024, n_eval: int = 256, vocab: int = 30522):
    g = torch.Generator().manual_seed(0)

    def synth(n):
        ids = torch.randint(0, vocab, (n, seq_len), generator=g)
        mask = torch.ones(n, seq_len, dtype=torch.long)
        types = torch.zeros(n, seq_len, dtype=torch.long)
        labels = torch.randint(0, 2, (n,), generator=g)
        return TensorDataset(ids, mask, types, labels)

    train = DataLoader(synth(n_train), shuffle=True, batch_size=batch_size, drop_last=True)
    evald = DataLoader(synth(n_eval), shuffle=False, batch_size=EVAL_BATCH_SIZE)
    return train, evald


def training_function(config, args):
    accelerator = Accelerator(mixed_precision=args.mixed_precision, gradient_accumulation_steps=1)
    set_seed(config["seed"])
    batch_size = int(config["batch_size"])
    gradient_accumulation_steps = 1
    if batch_size > MAX_GPU_BATCH_SIZE:
        gradient_accumulation_steps = batch_size // MAX_GPU_BATCH_SIZE
        batch_size = MAX_GPU_BATCH_SIZE
    accelerator.gradient_accumulation_steps = gradient_accumulation_steps

    train_dl, eval_dl = get_dataloaders(batch_size)
    model = BertForSequenceClassification(BertConfig.bert_base())
    optimizer = torch.optim.AdamW(model.parameters(), lr=config["lr"])
    scheduler = torch.optim.lr_scheduler.LambdaLR(optimizer, lambda s: min(1.0, (s + 1) / 100))

    model, optimizer, train_dl, eval_dl, scheduler = accelerator.prepare(
        model, optimizer, train_dl, eval_dl, scheduler
    )

    for epoch in range(config["num_epochs"]):
        model.train()
        for batch in train_dl:
            with accelerator.accumulate(model):
                ids, mask, types, labels = batch
                out = model(ids, attention_mask=mask, token_type_ids=types, labels=labels)
                accelerator.backward(out["loss"])
                optimizer.step()
                scheduler.step()
                optimizer.zero_grad()

        model.eval()
        correct = total = 0
        for batch in eval_dl:
            ids, mask, types, labels = batch
            with torch.no_grad():
                out = model(ids, attention_mask=mask, token_type_ids=types)
            preds = out["logits"].argmax(-1)
            preds, labels = accelerator.gather_for_metrics((preds, labels))
            correct += (preds == labels).sum().item()
            total += labels.numel()
        accelerator.print(f"epoch {epoch}: eval accuracy {correct / max(total,1):.3f}")
    accelerator.end_training()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--mixed_precision", default=None, choices=["no", "fp16", "bf16", "fp8"])
    parser.add_argument("--cpu", action="store_true")
    args = parser.parse_args()
    config = {"lr": 2e-5, "num_epochs": 1, "seed": 42, "batch_size": 16}
    training_function(config, args)


if __name__ == "__main__":
    main()
