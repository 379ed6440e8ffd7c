"""Image-classification training loop (the reference's examples/cv_example.py
shape, offline: a compact CNN on synthetic images instead of timm+pets).

Shows the same Accelerator API is model-agnostic — nothing in prepare/
backward/gather is transformer-specific.

Run:
  python examples/cv_example.py
  python -m accelerate_amd launch --num_processes 8 examples/cv_example.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from accelerate_amd import Accelerator, set_seed


class SmallConvNet(nn.Module):
    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, 32, 3, stride=2, padding=1), nn.BatchNorm2d(32), nn.ReLU(),
            nn.Conv2d(32, 64, 3, stride=2, padding=1), nn.BatchNorm2d(64), nn.ReLU(),
            nn.Conv2d(64, 128, 3, stride=2, padding=1), nn.BatchNorm2d(128), nn.ReLU(),
            nn.AdaptiveAvgPool2d(1),
        )
        self.classifier = nn.Linear(128, num_classes)

    def forward(self, x):
        return self.classifier(self.features(x).flatten(1))


def get_dataloaders(batch_size: int, image_size: int = 64, n_train: int = 512, n_eval: int = 128):
    g = torch.Generator().manual_seed(0)

    def synth(n):
        images = torch.randn(n, 3, image_size, image_size, generator=g)
        labels = torch.randint(0, 10, (n,), generator=g)
        return TensorDataset(images, labels)

    train = DataLoader(synth(n_train), batch_size=batch_size, shuffle=True, drop_last=True)
    evald = DataLoader(synth(n_eval), batch_size=batch_size)
    return train, evald


def training_function(args):
    accelerator = Accelerator(mixed_precision=args.mixed_precision, cpu=args.cpu)
    set_seed(args.seed)
    train_dl, eval_dl = get_dataloaders(args.batch_size)
    model = SmallConvNet()
    optimizer = torch.optim.AdamW(model.parameters(), lr=args.lr)
    scheduler = torch.optim.lr_scheduler.OneCycleLR(
        optimizer, max_lr=args.lr, total_steps=args.epochs * len(train_dl)
    )
    model, optimizer, train_dl, eval_dl, scheduler = accelerator.prepare(
        model, optimizer, train_dl, eval_dl, scheduler
    )

    loss_fn = nn.CrossEntropyLoss()
    for epoch in range(args.epochs):
        model.train()
        for images, labels in train_dl:
            optimizer.zero_grad()
            loss = loss_fn(model(images), labels)
            accelerator.backward(loss)
            optimizer.step()
            scheduler.step()

        model.eval()
        correct = total = 0
        for images, labels in eval_dl:
            with torch.no_grad():
                preds = model(images).argmax(-1)
            preds, labels = accelerator.gather_for_metrics((preds, labels))
            correct += (preds == labels).sum().item()
            total += labels.numel()
        accelerator.print(f"epoch {epoch}: eval accuracy {correct / total:.3f} ({total} samples)")

    accelerator.end_training()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--mixed_precision", default=None, choices=[None, "no", "fp16", "bf16"])
    parser.add_argument("--cpu", action="store_true")
    parser.add_argument("--epochs", type=int, default=2)
    parser.add_argument("--batch_size", type=int, default=32)
    parser.add_argument("--lr", type=float, default=3e-3)
    parser.add_argument("--seed", type=int, default=0)
    training_function(parser.parse_args())


if __name__ == "__main__":
    main()
