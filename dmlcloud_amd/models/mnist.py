"""MNIST CNN — the headline benchmark model.

Same architecture as the reference example (reference
examples/mnist.py:27-37): Conv(1->16,3x3) -> ReLU -> MaxPool ->
Conv(16->16,3x3) -> ReLU -> MaxPool -> Flatten -> Linear(784,10).
"""

import torch
from torch import nn


def mnist_cnn() -> nn.Module:
    return nn.Sequential(
        nn.Conv2d(1, 16, 3, padding=1),
        nn.ReLU(),
        nn.MaxPool2d(2),
        nn.Conv2d(16, 16, 3, padding=1),
        nn.ReLU(),
        nn.MaxPool2d(2),
        nn.Flatten(),
        nn.Linear(784, 10),
    )


class SyntheticMnist(torch.utils.data.Dataset):
    """Random MNIST-shaped data (no network access for the real dataset)."""

    def __init__(self, n: int = 8192, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.images = torch.randn(n, 1, 28, 28, generator=g)
        self.labels = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        return self.images[idx], self.labels[idx]
