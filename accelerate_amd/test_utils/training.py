"""Universal numeric fixtures (reference: test_utils/training.py:22-47):
a 2-parameter linear model on synthetic y = 2x + 3 data."""

import numpy as np
import torch
from torch.utils.data import Dataset


class RegressionDataset(Dataset):
    def __init__(self, a=2, b=3, length=64, seed=None):
        rng = np.random.default_rng(seed)
        self.length = length
        self.x = rng.normal(size=(length,)).astype(np.float32)
        self.y = a * self.x + b + rng.normal(scale=0.1, size=(length,)).astype(np.float32)

    def __len__(self):
        return self.length

    def __getitem__(self, i):
        return {"x": self.x[i], "y": self.y[i]}


class RegressionModel(torch.nn.Module):
    def __init__(self, a=0, b=0, double_output=False):
        super().__init__()
        self.a = torch.nn.Parameter(torch.tensor(float(a)))
        self.b = torch.nn.Parameter(torch.tensor(float(b)))
        self.first_batch = True

    def forward(self, x=None):
        if self.first_batch:
            self.first_batch = False
        return x * self.a + self.b
