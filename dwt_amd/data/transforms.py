"""Minimal, dependency-light image transforms (no torchvision in this stack).

Covers exactly what the reference pipelines use
(usps_mnist.py:355-386, resnet50_dwt_mec_officehome.py:527-543):
Compose / Resize / RandomCrop / RandomHorizontalFlip / ToTensor /
Normalize / Lambda.  Input images are PIL Images or HWC numpy arrays.
"""
from __future__ import annotations

import random

import numpy as np
import torch

try:
    from PIL import Image
except ImportError:  # pragma: no cover
    Image = None


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class Lambda:
    def __init__(self, fn):
        self.fn = fn

    def __call__(self, x):
        return self.fn(x)


class Resize:
    """Resize to (h, w); accepts int (square) or tuple."""

    def __init__(self, size):
        self.size = (size, size) if isinstance(size, int) else tuple(size)

    def __call__(self, img):
        if isinstance(img, np.ndarray):
            img = Image.fromarray(self._to_uint8(img).squeeze())
        return img.resize((self.size[1], self.size[0]), Image.BILINEAR)

    @staticmethod
    def _to_uint8(arr):
        if arr.dtype == np.uint8:
            return arr
        return (np.clip(arr, 0.0, 1.0) * 255).astype(np.uint8)


class RandomCrop:
    def __init__(self, size):
        self.size = (size, size) if isinstance(size, int) else tuple(size)

    def __call__(self, img):
        th, tw = self.size
        if isinstance(img, np.ndarray):
            h, w = img.shape[:2]
            i = random.randint(0, h - th)
            j = random.randint(0, w - tw)
            return img[i:i + th, j:j + tw]
        w, h = img.size
        i = random.randint(0, h - th)
        j = random.randint(0, w - tw)
        return img.crop((j, i, j + tw, i + th))


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            if isinstance(img, np.ndarray):
                return np.ascontiguousarray(img[:, ::-1])
            return img.transpose(Image.FLIP_LEFT_RIGHT)
        return img


class ToTensor:
    """PIL/HWC-numpy -> CHW float tensor in [0, 1]."""

    def __call__(self, img):
        if isinstance(img, torch.Tensor):
            return img
        if Image is not None and isinstance(img, Image.Image):
            arr = np.asarray(img)
        else:
            arr = np.asarray(img)
        if arr.ndim == 2:
            arr = arr[:, :, None]
        t = torch.from_numpy(np.ascontiguousarray(arr.transpose(2, 0, 1)))
        if t.dtype == torch.uint8:
            return t.float().div_(255.0)
        return t.float()


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, t):
        return (t - self.mean) / self.std
