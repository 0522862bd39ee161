"""Synthetic MNIST-like dataset (no downloads offline): K classes of noisy
template digits rendered as 1x28x28 float tensors."""
import numpy as np
import torch
from torch.utils.data import Dataset


class SyntheticDigits(Dataset):

    def __init__(self, n: int = 2048, num_classes: int = 10, seed: int = 0):
        rng = np.random.RandomState(seed)
        self.templates = rng.rand(num_classes, 28, 28).astype(np.float32)
        self.labels = rng.randint(0, num_classes, size=n)
        noise = rng.randn(n, 28, 28).astype(np.float32) * 0.3
        self.images = self.templates[self.labels] + noise

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, i):
        return torch.from_numpy(self.images[i]).unsqueeze(0), int(self.labels[i])
