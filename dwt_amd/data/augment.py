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


# ---------------------------------------------------------------------------
# Batched, device-agnostic version (per-image CPU augmentation
# can starve 8 GPUs — this one runs the whole target batch on the GPU).
# ---------------------------------------------------------------------------

def affine_augment_batch(x: torch.Tensor,
                         mats: torch.Tensor = None,
                         std: float = 0.1,
                         generator: torch.Generator = None) -> torch.Tensor:
    """Batched equivalent of ``random_affine_augmentation``: x (N, C, H, W)
    on any device; per-image 2x2 matrix = I + N(0, std) entries, applied
    origin-anchored exactly like the scalar scipy path (out[o] = in[inv @ o]
    in pixel index space, zeros outside).  ``mats`` overrides the random
    matrices (N, 2, 2), acting on (row, col) indices like the scipy path."""
    n, _, h, w = x.shape
    dev = x.device
    if mats is None:
        mats = torch.eye(2, device=dev).expand(n, 2, 2).clone()
        mats = mats + std * torch.randn(n, 2, 2, device=dev,
                                        generator=generator)
    inv = torch.linalg.inv(mats.to(torch.float64)).to(torch.float32)  # (N,2,2)
    ys = torch.arange(h, device=dev, dtype=torch.float32)
    xs = torch.arange(w, device=dev, dtype=torch.float32)
    gy, gx = torch.meshgrid(ys, xs, indexing="ij")       # (H, W) row/col
    o = torch.stack([gy, gx])                            # (2, H, W)
    # p = inv @ o per image: (N, 2, H, W)
    p = torch.einsum("nij,jhw->nihw", inv, o)
    grid = torch.empty(n, h, w, 2, device=dev, dtype=torch.float32)
    grid[..., 0] = 2.0 * p[:, 1] / max(w - 1, 1) - 1.0   # x (col)
    grid[..., 1] = 2.0 * p[:, 0] / max(h - 1, 1) - 1.0   # y (row)
    return torch.nn.functional.grid_sample(
        x.float(), grid, mode="bilinear", padding_mode="zeros",
        align_corners=True).to(x.dtype)


def gpu_target_views(x: torch.Tensor, std: float = 0.1, flip_p: float = 0.5,
                     generator: torch.Generator = None) -> torch.Tensor:
    """Build the MEC duplicate target view on-device: random horizontal flip
    + the random affine (the reference's blur is a no-op at its sigma).
    Opt-in (--gpu_augment): the affine runs AFTER normalization here, so
    out-of-frame pixels are 0 in normalized space rather than the
    reference's normalized-zero constant — a border-only deviation."""
    n = x.shape[0]
    flip = torch.rand(n, device=x.device, generator=generator) < flip_p
    out = torch.where(flip[:, None, None, None], torch.flip(x, dims=[-1]), x)
    return affine_augment_batch(out, std=std, generator=generator)
