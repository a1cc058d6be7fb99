"""Synthetic data (there is no dataset/network access in this environment).

``SyntheticImages`` is fully deterministic per index (so the push
re-forward of a chosen image reproduces pass-1 features exactly), and is
shaped like the CUB configs (224x224, C classes).

``DeviceBatchPool`` pre-generates a handful of batches directly in device
memory for benchmarking — the bench measures the training step, not host
RNG.
"""

import torch
from torch.utils.data import Dataset


class SyntheticImages(Dataset):
    """Deterministic random images. Returns (image, label, index)."""

    def __init__(self, n: int, num_classes: int, img_size: int = 224,
                 seed: int = 0, normalize: bool = True):
        self.n = n
        self.num_classes = num_classes
        self.img_size = img_size
        self.seed = seed
        self.normalize = normalize

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 1000003 + i)
        img = torch.randn(3, self.img_size, self.img_size, generator=g)
        if not self.normalize:
            img = torch.sigmoid(img)  # push loaders expect [0,1] images
        label = i % self.num_classes
        return img, label, i


class DeviceBatchPool:
    """A rotating pool of pre-generated device batches (bench input)."""

    def __init__(self, batch_size: int, num_classes: int, img_size: int = 224,
                 device='cuda', pool: int = 4, seed: int = 0,
                 dtype: torch.dtype = torch.float32,
                 channels_last: bool = False):
        g = torch.Generator(device='cpu').manual_seed(seed)
        self.images = []
        self.labels = []
        for i in range(pool):
            img = torch.randn(batch_size, 3, img_size, img_size, generator=g,
                              dtype=dtype)
            lab = torch.randint(0, num_classes, (batch_size,), generator=g)
            img = img.to(device)
            if channels_last:
                img = img.contiguous(memory_format=torch.channels_last)
            self.images.append(img)
            self.labels.append(lab.to(device))
        self._i = 0

    def next(self):
        b = (self.images[self._i], self.labels[self._i])
        self._i = (self._i + 1) % len(self.images)
        return b
