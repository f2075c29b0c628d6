"""USPS digits dataset (usps_28x28.pkl gzip-pickle layout).

Behavior mirrors the reference loader (usps_mnist.py:26-120): the train split
is oversampled x6 to balance against MNIST's 60k, images
come out as HWC float arrays for the transform pipeline, and the dataset
supports a dual-transform mode returning ``(img, img_aug, label)``.

There is no network in this environment: ``download=True`` is accepted for
CLI parity but only verifies the file exists.
"""
from __future__ import annotations

import gzip
import os
import pickle

import numpy as np
import torch
import torch.utils.data as data

USPS_TRAIN_MULTIPLIER = 6


class USPS(data.Dataset):
    filename = "usps_28x28.pkl"

    def __init__(self, root, train=True, transform=None, transform_aug=None,
                 download=False):
        self.root = os.path.expanduser(root)
        self.train = train
        self.transform = transform
        self.transform_aug = transform_aug

        path = os.path.join(self.root, self.filename)
        if not os.path.exists(path):
            raise RuntimeError(
                f"USPS pickle not found at {path}. This environment has no "
                "network; place usps_28x28.pkl there, or use --synthetic.")

        with gzip.open(path, "rb") as f:
            blob = pickle.load(f, encoding="bytes")
        split = blob[0] if train else blob[1]
        images, labels = split[0], split[1]
        self.dataset_size = labels.shape[0]

        if train:
            images = np.repeat(images, USPS_TRAIN_MULTIPLIER, axis=0)
            labels = np.repeat(labels, USPS_TRAIN_MULTIPLIER, axis=0)
            idx = np.arange(labels.shape[0])
            np.random.shuffle(idx)
            images, labels = images[idx], labels[idx]

        # NCHW float -> NHWC for the transform pipeline
        self.data = images.transpose(0, 2, 3, 1)
        self.labels = labels

    def __getitem__(self, index):
        img, label = self.data[index], self.labels[index]
        img_aug = self.transform_aug(img) if self.transform_aug is not None else None
        if self.transform is not None:
            img = self.transform(img)
        label = torch.as_tensor(int(label), dtype=torch.long)
        if img_aug is not None:
            return img, img_aug, label
        return img, label

    def __len__(self):
        return self.labels.shape[0]
