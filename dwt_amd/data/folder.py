"""Class-per-directory image dataset with dual-transform output.

Fresh implementation of the capability of the reference's torchvision-fork
``utils/folder.py:58-218``: scan ``root/<class>/<image>``, map classes to
indices alphabetically, and — when ``transform_aug`` is given — return
``(img, img_aug, label)`` so the Office-Home target loader yields the
augmented duplicate stream for the MEC loss.
"""
from __future__ import annotations

import os
from typing import Callable, List, Optional, Tuple

import torch.utils.data as data

try:
    from PIL import Image
except ImportError:  # pragma: no cover
    Image = None

IMG_EXTENSIONS = (".jpg", ".jpeg", ".png", ".ppm", ".bmp", ".pgm", ".tif",
                  ".tiff", ".webp")


def pil_loader(path: str):
    with open(path, "rb") as f:
        img = Image.open(f)
        return img.convert("RGB")


def has_file_allowed_extension(filename: str, extensions) -> bool:
    return filename.lower().endswith(tuple(extensions))


def make_dataset(directory: str, class_to_idx, extensions) -> List[Tuple[str, int]]:
    samples = []
    directory = os.path.expanduser(directory)
    for cls in sorted(class_to_idx.keys()):
        d = os.path.join(directory, cls)
        if not os.path.isdir(d):
            continue
        for root, _, fnames in sorted(os.walk(d)):
            for fname in sorted(fnames):
                if has_file_allowed_extension(fname, extensions):
                    samples.append((os.path.join(root, fname), class_to_idx[cls]))
    return samples


class DatasetFolder(data.Dataset):
    def __init__(self, root: str, loader: Callable, extensions,
                 transform: Optional[Callable] = None,
                 transform_aug: Optional[Callable] = None,
                 target_transform: Optional[Callable] = None):
        self.root = root
        classes = sorted(e.name for e in os.scandir(os.path.expanduser(root))
                         if e.is_dir())
        if not classes:
            raise RuntimeError(f"no class directories found under {root}")
        self.classes = classes
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = make_dataset(root, self.class_to_idx, extensions)
        if not self.samples:
            raise RuntimeError(f"no images with extensions {extensions} under {root}")
        self.loader = loader
        self.transform = transform
        self.transform_aug = transform_aug
        self.target_transform = target_transform
        self.targets = [s[1] for s in self.samples]

    def __getitem__(self, index):
        path, target = self.samples[index]
        sample = self.loader(path)
        aug = self.transform_aug(sample) if self.transform_aug is not None else None
        if self.transform is not None:
            sample = self.transform(sample)
        if self.target_transform is not None:
            target = self.target_transform(target)
        if aug is not None:
            return sample, aug, target
        return sample, target

    def __len__(self):
        return len(self.samples)


class ImageFolder(DatasetFolder):
    def __init__(self, root, transform=None, transform_aug=None,
                 target_transform=None, loader=pil_loader):
        super().__init__(root, loader, IMG_EXTENSIONS, transform=transform,
                         transform_aug=transform_aug,
                         target_transform=target_transform)
        self.imgs = self.samples
