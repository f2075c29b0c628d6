"""Data pipeline: transforms, dataset formats, dual-transform contract."""
import gzip
import os
import pickle

import numpy as np
import pytest
import torch

from dwt_amd.data import (MNIST, USPS, Compose, ImageFolder, Normalize,
                          RandomCrop, RandomHorizontalFlip, Resize,
                          SyntheticDigits, SyntheticOfficeHome, ToTensor)
from dwt_amd.data.augment import gaussian_blur, random_affine_augmentation


def test_transforms_pipeline():
    from PIL import Image
    img = Image.fromarray((np.random.rand(40, 50, 3) * 255).astype(np.uint8))
    t = Compose([Resize((32, 32)), RandomCrop(28), RandomHorizontalFlip(),
                 ToTensor(), Normalize([0.5] * 3, [0.5] * 3)])
    out = t(img)
    assert out.shape == (3, 28, 28)
    assert out.dtype == torch.float32
    assert out.min() >= -1.001 and out.max() <= 1.001


def test_to_tensor_numpy_hwc():
    arr = np.random.rand(28, 28, 1).astype(np.float32)
    out = ToTensor()(arr)
    assert out.shape == (1, 28, 28)
    assert torch.allclose(out[0], torch.from_numpy(arr[:, :, 0]))


def test_usps_pickle_roundtrip(tmp_path):
    root = tmp_path / "usps"
    root.mkdir()
    train = (np.random.rand(11, 1, 28, 28).astype(np.float32),
             np.random.randint(0, 10, 11))
    testsp = (np.random.rand(5, 1, 28, 28).astype(np.float32),
              np.random.randint(0, 10, 5))
    with gzip.open(root / "usps_28x28.pkl", "wb") as f:
        pickle.dump([train, testsp], f)
    t = Compose([ToTensor(), Normalize([0.5], [0.5])])
    ds = USPS(str(root), train=True, transform=t)
    assert len(ds) == 66  # x6 oversampling
    img, label = ds[0]
    assert img.shape == (1, 28, 28) and label.dtype == torch.long
    ds_test = USPS(str(root), train=False, transform=t)
    assert len(ds_test) == 5


def test_mnist_pt_roundtrip(tmp_path):
    root = tmp_path / "mnist"
    (root / "processed").mkdir(parents=True)
    data = torch.randint(0, 255, (7, 28, 28), dtype=torch.uint8)
    targets = torch.randint(0, 10, (7,))
    torch.save((data, targets), root / "processed" / "training.pt")
    t = Compose([ToTensor(), Normalize([0.1307], [0.3081])])
    ds = MNIST(str(root), train=True, transform=t, transform_aug=t)
    img, aug, label = ds[0]   # dual-transform contract
    assert img.shape == (1, 28, 28) and aug.shape == (1, 28, 28)


def test_image_folder_dual_transform(tmp_path):
    from PIL import Image
    for cls in ("alpha", "beta"):
        d = tmp_path / cls
        d.mkdir()
        for i in range(3):
            Image.fromarray((np.random.rand(30, 30, 3) * 255).astype(np.uint8)) \
                .save(d / f"img{i}.jpg")
    t = Compose([Resize((16, 16)), ToTensor()])
    ds = ImageFolder(str(tmp_path), transform=t)
    assert len(ds) == 6
    assert ds.class_to_idx == {"alpha": 0, "beta": 1}
    img, label = ds[0]
    assert img.shape == (3, 16, 16) and label == 0
    ds2 = ImageFolder(str(tmp_path), transform=t, transform_aug=t)
    img, aug, label = ds2[0]
    assert aug.shape == (3, 16, 16)


def test_augmentations():
    x = torch.rand(3, 32, 32)
    y = random_affine_augmentation(x)
    assert y.shape == x.shape and not torch.equal(x, y)
    # sigma=0.1 -> ksize 1 -> identity
    assert torch.equal(gaussian_blur(x, 0.1), x)
    assert not torch.equal(gaussian_blur(x, 2.0), x)


def test_synthetic_datasets():
    ds = SyntheticDigits(32)
    img, label = ds[0]
    assert img.shape == (1, 28, 28)
    oh = SyntheticOfficeHome(16, img_size=64, transform_aug=True)
    img, aug, label = oh[3]
    assert img.shape == (3, 64, 64) and aug.shape == (3, 64, 64)
    assert not torch.equal(img, aug)
    img2, _, label2 = oh[3]
    assert torch.equal(img, img2) and label == label2  # deterministic per index


def test_resize_numpy_path():
    arr = np.random.rand(30, 40, 1).astype(np.float32)
    out = Resize((16, 16))(arr)
    t = ToTensor()(out)
    assert t.shape == (1, 16, 16)
    rgb = (np.random.rand(20, 20, 3) * 255).astype(np.uint8)
    out = Resize(10)(rgb)
    assert ToTensor()(out).shape == (3, 10, 10)


def test_affine_augment_batch_matches_scalar_path():
    """The batched grid_sample version must agree with the per-image scipy
    path given the same matrix (interior pixels; border bilinear handling
    differs between the two libraries)."""
    import numpy as np
    import torch
    from dwt_amd.data import augment

    torch.manual_seed(0)
    np.random.seed(0)
    h = w = 24
    # smooth test image so interpolation differences stay tiny
    yy, xx = np.meshgrid(np.linspace(-1, 1, h), np.linspace(-1, 1, w),
                         indexing="ij")
    img = np.exp(-(yy ** 2 + xx ** 2) * 3).astype(np.float32)
    x = torch.from_numpy(img)[None].repeat(3, 1, 1)  # CHW

    mat = np.float32([[1.05, 0.08], [-0.06, 0.97]])
    # scalar path with a pinned matrix (reproduce its internals)
    inv = np.linalg.inv(mat.astype(np.float64))
    from scipy import ndimage
    ref = np.stack([ndimage.affine_transform(x[c].numpy(), inv, order=1,
                                             mode="constant", cval=0.0)
                    for c in range(3)])

    out = augment.affine_augment_batch(
        x[None], mats=torch.from_numpy(mat)[None])[0]
    diff = (out.numpy() - ref)[:, 4:-4, 4:-4]
    assert np.abs(diff).max() < 2e-3, np.abs(diff).max()

    # random mode: shape/device/dtype preserved, identity at std=0
    b = torch.randn(4, 3, 16, 16)
    same = augment.affine_augment_batch(b, std=0.0)
    assert torch.allclose(same, b, atol=1e-5)
    r = augment.affine_augment_batch(b, std=0.1)
    assert r.shape == b.shape and r.dtype == b.dtype
    assert not torch.allclose(r, b)
