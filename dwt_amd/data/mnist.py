"""MNIST dataset reading the torchvision `processed/{training,test}.pt` layout
(usps_mnist.py:123-181), with optional dual-transform output."""
from __future__ import annotations

import os

import torch
import torch.utils.data as data

try:
    from PIL import Image
except ImportError:  # pragma: no cover
    Image = None


class MNIST(data.Dataset):
    processed_folder = "processed"
    training_file = "training.pt"
    test_file = "test.pt"

    def __init__(self, root, train=True, transform=None, transform_aug=None):
        self.root = os.path.expanduser(root)
        self.transform = transform
        self.transform_aug = transform_aug
        self.train = train
        data_file = self.training_file if train else self.test_file
        path = os.path.join(self.root, self.processed_folder, data_file)
        if not os.path.exists(path):
            raise RuntimeError(
                f"MNIST data not found at {path}. This environment has no "
                "network; place processed/{{training,test}}.pt there, or use "
                "--synthetic.")
        self.data, self.targets = torch.load(path, weights_only=False)

    def __getitem__(self, index):
        img, target = self.data[index], self.targets[index]
        pil = Image.fromarray(img.numpy(), mode="L")
        img_aug = self.transform_aug(pil) if self.transform_aug is not None else None
        out = self.transform(pil) if self.transform is not None else pil
        if img_aug is not None:
            return out, img_aug, target
        return out, target

    def __len__(self):
        return len(self.data)
