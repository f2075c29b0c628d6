"""Synthetic datasets for offline / benchmark runs (no network, no datasets
on disk — BASELINE.json mandates synthetic data + random-init weights for the
perf numbers).

The digits dataset is *learnable* (class-dependent structure), so the CPU and
single-GPU smoke trainings can demonstrate decreasing loss / above-chance
accuracy, not just throughput.
"""
from __future__ import annotations

import numpy as np
import torch
import torch.utils.data as data


class SyntheticDigits(data.Dataset):
    """28x28 single-channel 10-class dataset: class-dependent frequency
    gratings + noise.  Dual-transform mode mirrors USPS/MNIST loaders."""

    def __init__(self, n=512, train=True, transform=None, transform_aug=None,
                 seed=0, shift=0.0):
        rng = np.random.RandomState(seed + (0 if train else 1))
        self.labels = rng.randint(0, 10, size=n)
        xs = np.linspace(0, 1, 28)
        xx, yy = np.meshgrid(xs, xs)
        imgs = np.empty((n, 28, 28, 1), dtype=np.float32)
        for i, lab in enumerate(self.labels):
            freq = 1.5 + 0.7 * lab
            phase = rng.uniform(0, 2 * np.pi)
            img = 0.5 + 0.4 * np.sin(freq * np.pi * (xx + 0.3 * lab * yy) + phase)
            img += shift + rng.normal(0, 0.08, size=img.shape)
            imgs[i, :, :, 0] = np.clip(img, 0, 1)
        self.data = imgs
        self.transform = transform
        self.transform_aug = transform_aug

    def __getitem__(self, index):
        img, label = self.data[index], self.labels[index]
        aug = self.transform_aug(img) if self.transform_aug is not None else None
        if self.transform is not None:
            img = self.transform(img)
        else:
            img = torch.from_numpy(img.transpose(2, 0, 1))
        label = torch.as_tensor(int(label), dtype=torch.long)
        if aug is not None:
            return img, aug, label
        return img, label

    def __len__(self):
        return len(self.labels)


class SyntheticOfficeHome(data.Dataset):
    """3x224x224 65-class random-image dataset shaped like Office-Home.

    Generated on the fly (deterministic per index) so a benchmark-size
    dataset costs no memory.
    """

    def __init__(self, n=2048, num_classes=65, img_size=224, transform=None,
                 transform_aug=None, seed=0):
        self.n = n
        self.num_classes = num_classes
        self.img_size = img_size
        self.transform = transform
        self.transform_aug = transform_aug
        self.seed = seed

    def _sample(self, index):
        g = torch.Generator().manual_seed(self.seed * 1000003 + index)
        label = int(torch.randint(0, self.num_classes, (1,), generator=g))
        img = torch.randn(3, self.img_size, self.img_size, generator=g) * 0.5
        img[0] += (label / self.num_classes) - 0.5
        return img, label

    def __getitem__(self, index):
        img, label = self._sample(index)
        aug = None
        if self.transform_aug is not None:
            if callable(self.transform_aug):
                aug = self.transform_aug(img)
            else:  # default "augmented view": jittered copy
                g = torch.Generator().manual_seed(self.seed * 2000003 + index)
                aug = img + torch.randn(img.shape, generator=g) * 0.05
        if self.transform is not None and callable(self.transform):
            img = self.transform(img)
        label = torch.as_tensor(label, dtype=torch.long)
        if aug is not None:
            return img, aug, label
        return img, label

    def __len__(self):
        return self.n
