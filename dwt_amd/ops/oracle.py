"""Pure-PyTorch oracle of the DWT / domain-BN / loss math.

This module is the semantic ground truth for the whole framework: every HIP
kernel and every hand-written backward is numerically validated against the
functions here (in fp32/fp64).  The math reproduces the reference behavior
documented in SURVEY.md:

* grouped whitening: /root/reference/utils/whitening.py:37-61
* domain BN:         /root/reference/utils/batch_norm.py:54-69
* MEC loss:          /root/reference/utils/consensus_loss.py:11-24
* entropy loss:      /root/reference/usps_mnist.py:188-194

Everything is differentiable through torch autograd, so the explicit
backwards in `functional.py` can be gradchecked against these.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

# ----------------------------------------------------------------------------
# Grouped statistics
# ----------------------------------------------------------------------------


def channel_mean(x: torch.Tensor) -> torch.Tensor:
    """Per-channel mean over (N, H, W) of an NCHW tensor -> shape (1, C, 1, 1)."""
    return x.mean(dim=(0, 2, 3), keepdim=True)


def grouped_cov(xn: torch.Tensor, num_groups: int) -> torch.Tensor:
    """Per-group covariance of a centered NCHW tensor.

    Treats each group's ``g`` channels as variables and all N*H*W positions as
    observations: ``cov[G] = T_G @ T_G.T / M`` with ``T`` of shape (G, g, M).
    (reference math: whitening.py:46-47)
    """
    n, c, h, w = xn.shape
    g = c // num_groups
    t = xn.permute(1, 0, 2, 3).reshape(num_groups, g, n * h * w)
    return torch.bmm(t, t.transpose(1, 2)) / t.shape[-1]


def shrink_cov(cov: torch.Tensor, eps: float) -> torch.Tensor:
    """Shrinkage toward identity: (1-eps) * cov + eps * I (whitening.py:48)."""
    g = cov.shape[-1]
    eye = torch.eye(g, dtype=cov.dtype, device=cov.device)
    return (1.0 - eps) * cov + eps * eye


# ----------------------------------------------------------------------------
# Whitening matrices
# ----------------------------------------------------------------------------


def whiten_matrix_chol(cov_shrunk: torch.Tensor) -> torch.Tensor:
    """Triangular (Cholesky) whitening matrix ``W = L^{-1}``, ``A = L L^T``.

    Reference-parity mode (whitening.py:53).  Batched over groups: input
    (G, g, g) SPD, output (G, g, g) lower-triangular with
    ``W @ A @ W.T == I``.
    """
    ell = torch.linalg.cholesky(cov_shrunk)
    return torch.linalg.solve_triangular(
        ell, torch.eye(ell.shape[-1], dtype=ell.dtype, device=ell.device).expand_as(ell),
        upper=False,
    )


def whiten_matrix_ns(cov_shrunk: torch.Tensor, iters: int = 7) -> torch.Tensor:
    """ZCA whitening matrix ``W = A^{-1/2}`` via Newton–Schulz iteration.

    MI355X-native primary mode: symmetric, iteration-only (no solver), maps to
    a handful of tiny batched matmuls on the GPU and has a mechanical unrolled
    backward.  Normalizes by trace so the iteration converges for any SPD A.
    """
    g = cov_shrunk.shape[-1]
    eye = torch.eye(g, dtype=cov_shrunk.dtype, device=cov_shrunk.device).expand_as(cov_shrunk)
    # trace normalization: s = tr(A), guaranteed >= g*eps > 0 after shrinkage
    s = cov_shrunk.diagonal(dim1=-2, dim2=-1).sum(-1).clamp_min(1e-30)
    s_ = s.view(-1, 1, 1)
    y = cov_shrunk / s_
    z = eye.clone()
    for _ in range(iters):
        t = 0.5 * (3.0 * eye - torch.bmm(z, y))
        y = torch.bmm(y, t)
        z = torch.bmm(t, z)
    return z / torch.sqrt(s_)


def whitening_forward(
    x: torch.Tensor,
    num_groups: int,
    eps: float = 1e-3,
    mode: str = "chol",
    running_mean: torch.Tensor | None = None,
    running_var: torch.Tensor | None = None,
    training: bool = True,
    ns_iters: int = 7,
):
    """Full whitening forward.

    Training: batch stats (mean over N,H,W; per-group cov), shrink, whiten.
    Eval (running stats given): use running mean and *re-shrunk* running cov
    (whitening.py:42-43, 50-51 — the EMA stores the unshrunk cov).

    Returns (y, mean, cov_batch) — cov_batch is the *unshrunk* covariance used
    for the EMA update; in eval mode it is None (the reference wastefully
    computes it; we skip — SURVEY quirk #5).
    """
    n, c, h, w = x.shape
    g = c // num_groups
    if training or running_mean is None:
        m = channel_mean(x)
        xn = x - m
        cov = grouped_cov(xn, num_groups)
        cov_s = shrink_cov(cov, eps)
    else:
        m = running_mean.reshape(1, c, 1, 1).to(x.dtype)
        xn = x - m
        cov = None
        cov_s = shrink_cov(running_var.to(x.dtype), eps)

    wmat = whiten_matrix_chol(cov_s) if mode == "chol" else whiten_matrix_ns(cov_s, ns_iters)
    weight = wmat.reshape(c, g, 1, 1)
    y = F.conv2d(xn, weight, groups=num_groups)
    return y, m, cov


# ----------------------------------------------------------------------------
# Domain BatchNorm (stock semantics, injected running buffers)
# ----------------------------------------------------------------------------


def batch_norm_forward(
    x: torch.Tensor,
    running_mean: torch.Tensor | None,
    running_var: torch.Tensor | None,
    weight: torch.Tensor | None,
    bias: torch.Tensor | None,
    use_batch_stats: bool,
    momentum: float,
    eps: float = 1e-5,
):
    """Stock F.batch_norm semantics (batch_norm.py:66-69): biased variance for
    normalization, unbiased for the EMA, running = (1-m)*running + m*batch."""
    return F.batch_norm(x, running_mean, running_var, weight, bias,
                        use_batch_stats, momentum, eps)


# ----------------------------------------------------------------------------
# Losses
# ----------------------------------------------------------------------------


def mec_loss(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Min-Entropy Consensus loss (consensus_loss.py:11-24).

    For two views' logits: ``mean_n min_k -1/2 (log p_x[k] + log p_y[k])``.
    """
    lx = F.log_softmax(x, dim=1)
    ly = F.log_softmax(y, dim=1)
    return (-0.5 * (lx + ly)).min(dim=1).values.mean()


def entropy_loss(x: torch.Tensor) -> torch.Tensor:
    """Shannon entropy of the softmax distribution (usps_mnist.py:188-194)."""
    p = F.softmax(x, dim=1)
    q = F.log_softmax(x, dim=1)
    return -(p * q).sum(dim=-1).mean()


def source_ce_loss(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """nll_loss(log_softmax(x)) as both reference train loops use
    (usps_mnist.py:298, resnet50_dwt_mec_officehome.py:425)."""
    return F.nll_loss(F.log_softmax(logits, dim=1), target)
