"""Target-stream augmentations for the MEC duplicate view.

Equivalents of the reference's cv2-based augs
(resnet50_dwt_mec_officehome.py:481-492) built on scipy.ndimage (no cv2 in
this stack):

* ``random_affine_augmentation``: identity matrix perturbed by N(0, 0.1)
  entries applied to a CHW tensor.
* ``gaussian_blur``: the reference's sigma=0.1 yields kernel size 1 — an
  identity op.  We keep the same ksize formula: blur only
  happens when sigma is large enough to matter.
"""
from __future__ import annotations

import numpy as np
import torch
from scipy import ndimage


def random_affine_augmentation(x: torch.Tensor) -> torch.Tensor:
    """x: CHW float tensor -> warped CHW tensor."""
    a = np.float32([[1 + np.random.normal(0.0, 0.1), np.random.normal(0.0, 0.1)],
                    [np.random.normal(0.0, 0.1), 1 + np.random.normal(0.0, 0.1)]])
    arr = x.numpy()
    # map output coords through the inverse of the forward matrix
    # (cv2.warpAffine semantics: dst(p) = src(M^{-1} p))
    inv = np.linalg.inv(a.astype(np.float64))
    out = np.empty_like(arr)
    for c in range(arr.shape[0]):
        out[c] = ndimage.affine_transform(arr[c], inv, order=1, mode="constant", cval=0.0)
    return torch.from_numpy(out)


def gaussian_blur(x: torch.Tensor, sigma: float = 0.1) -> torch.Tensor:
    ksize = int(sigma + 0.5) * 8 + 1
    if ksize <= 1:
        return x
    arr = x.numpy()
    out = np.empty_like(arr)
    for c in range(arr.shape[0]):
        out[c] = ndimage.gaussian_filter(arr[c], sigma=sigma, truncate=(ksize // 2) / max(sigma, 1e-6))
    return torch.from_numpy(out)


# reference-style aliases
_random_affine_augmentation = random_affine_augmentation
_gaussian_blur = gaussian_blur
