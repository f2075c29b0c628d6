"""Autograd functions for the DWT compute with explicit, hand-derived backwards.

These functions define the *algorithm contract* for the HIP kernels: the
gfx950 kernels in ``dwt_amd/kernels`` implement exactly the passes below
(stats -> matrix-function -> fused apply; reduce -> matrix-function backward
-> fused apply-backward).  The torch implementation here is used:

* on CPU (BASELINE config 1 mandates a CPU plumbing path), and
* as the reference the GPU kernels are numerically tested against.

On a ROCm device with the extension built, the Functions dispatch to the HIP
kernels (see ``dwt_amd/kernels``).

Math reproduced from the reference (not its code):
  whitening fwd: /root/reference/utils/whitening.py:37-61
  the backward is derived by hand (the reference relied on autograd through
  cholesky/inverse/bmm):

    y_G = W_G T_G          T_G = centered x, grouped (G, g, M)
    Sigma = T T^T / M      Sigma_s = (1-eps) Sigma + eps I
    W = chol(Sigma_s)^{-1}           (mode 'chol', reference parity)
    W = Sigma_s^{-1/2} by Newton-Schulz (mode 'zca', MI355X-primary)

  Given dY:
    dT_direct = W^T dY
    dW        = dY T^T
    dSigma_s  = matrix-function backward (per mode, below)
    dT_cov    = ((1-eps)/M) (dSigma_s + dSigma_s^T) T
    dT        = dT_direct + dT_cov
    dx        = dT - mean_{N,H,W}(dT)     (mean path, like BN backward)
"""
from __future__ import annotations

import os
from typing import List, Optional, Sequence

import torch
import torch.nn.functional as F

from . import oracle


# ----------------------------------------------------------------------------
# Matrix-function forward/backward (batched over groups, shapes (G, g, g))
# ----------------------------------------------------------------------------


def matfn_chol_forward(cov_s: torch.Tensor):
    """W = L^{-1}, L = chol(cov_s).  Returns (W, L)."""
    ell = torch.linalg.cholesky(cov_s)
    eye = torch.eye(ell.shape[-1], dtype=ell.dtype, device=ell.device).expand_as(ell)
    w = torch.linalg.solve_triangular(ell, eye, upper=False)
    return w, ell


def matfn_chol_backward(w: torch.Tensor, ell: torch.Tensor, w_bar: torch.Tensor):
    """d cov_s for W = chol(cov_s)^{-1}.

    L_bar   = tril(-W^T W_bar W^T)
    Phi     = tril(L^T L_bar) with halved diagonal
    A_bar   = sym(W^T Phi W)
    """
    l_bar = torch.tril(-w.transpose(-1, -2) @ w_bar @ w.transpose(-1, -2))
    s = ell.transpose(-1, -2) @ l_bar
    phi = torch.tril(s)
    phi = phi - 0.5 * torch.diag_embed(phi.diagonal(dim1=-2, dim2=-1))
    a_bar = w.transpose(-1, -2) @ phi @ w
    return 0.5 * (a_bar + a_bar.transpose(-1, -2))


def matfn_ns_forward(cov_s: torch.Tensor, iters: int):
    """W = cov_s^{-1/2} via trace-normalized Newton-Schulz.

    Returns (W, saved) where saved = (ys, zs, s): the per-iteration inputs
    needed for the unrolled backward.
    """
    g = cov_s.shape[-1]
    eye = torch.eye(g, dtype=cov_s.dtype, device=cov_s.device).expand_as(cov_s)
    s = cov_s.diagonal(dim1=-2, dim2=-1).sum(-1).clamp_min(1e-30).view(-1, 1, 1)
    y = cov_s / s
    z = eye.clone()
    ys: List[torch.Tensor] = []
    zs: List[torch.Tensor] = []
    for _ in range(iters):
        ys.append(y)
        zs.append(z)
        t = 0.5 * (3.0 * eye - torch.bmm(z, y))
        y = torch.bmm(y, t)
        z = torch.bmm(t, z)
    w = z / torch.sqrt(s)
    return w, (ys, zs, s, z)


def matfn_ns_backward(saved, w_bar: torch.Tensor):
    """Unrolled backward of the Newton-Schulz inverse-sqrt."""
    ys, zs, s, z_k = saved
    g = w_bar.shape[-1]
    eye = torch.eye(g, dtype=w_bar.dtype, device=w_bar.device).expand_as(w_bar)
    inv_sqrt_s = 1.0 / torch.sqrt(s)
    z_bar = w_bar * inv_sqrt_s
    # W = Z_K * s^{-1/2}
    s_bar = (w_bar * z_k).sum(dim=(-2, -1), keepdim=True) * (-0.5) * inv_sqrt_s / s
    y_bar = torch.zeros_like(w_bar)
    for y, z in zip(reversed(ys), reversed(zs)):
        t = 0.5 * (3.0 * eye - torch.bmm(z, y))
        t_bar = torch.bmm(y.transpose(-1, -2), y_bar) + torch.bmm(z_bar, z.transpose(-1, -2))
        y_bar = torch.bmm(y_bar, t.transpose(-1, -2)) - 0.5 * torch.bmm(z.transpose(-1, -2), t_bar)
        z_bar = torch.bmm(t.transpose(-1, -2), z_bar) - 0.5 * torch.bmm(t_bar, y.transpose(-1, -2))
    # y_0 = A / s ; s = tr(A)
    y0 = ys[0]
    s_bar = s_bar - (y_bar * y0).sum(dim=(-2, -1), keepdim=True) / s
    a_bar = y_bar / s + s_bar * eye
    return 0.5 * (a_bar + a_bar.transpose(-1, -2))


# ----------------------------------------------------------------------------
# Torch helpers shared by forward/backward
# ----------------------------------------------------------------------------


def _dist_world():
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
        return dist, dist.get_world_size()
    return None, 1


def _grouped_apply(xn: torch.Tensor, w: torch.Tensor, num_groups: int) -> torch.Tensor:
    """y[:, Gg+i] = sum_j W[G, i, j] xn[:, Gg+j] — the grouped 1x1 'conv'."""
    n, c, h, hw = xn.shape
    g = c // num_groups
    t = xn.reshape(n, num_groups, g, h * hw)
    y = torch.einsum("gij,ngjm->ngim", w, t)
    return y.reshape(n, c, h, hw)


def _grouped_outer(a: torch.Tensor, b: torch.Tensor, num_groups: int) -> torch.Tensor:
    """out[G] = sum over batch/space of a_G b_G^T: (G, g, g)."""
    n, c, h, w = a.shape
    g = c // num_groups
    ta = a.permute(1, 0, 2, 3).reshape(num_groups, g, n * h * w)
    tb = b.permute(1, 0, 2, 3).reshape(num_groups, g, n * h * w)
    return torch.bmm(ta, tb.transpose(1, 2))


# ----------------------------------------------------------------------------
# The fused multi-branch whitening op
# ----------------------------------------------------------------------------


class WhitenMulti(torch.autograd.Function):
    """Fused DWT site: split batch into `parts` domain branches, whiten each
    with its own statistics/EMA buffers, concatenate, then (optionally) apply
    a shared affine gamma/beta and ReLU.

    This is the exact structure of every norm site in the reference models
    (usps_mnist.py:237, resnet50_dwt_mec_officehome.py:221-222): whitening is
    per-domain, gamma/beta are shared across domains, ReLU follows (except on
    the pre-residual site).
    """

    @staticmethod
    def forward(ctx, x, gamma, beta, running_means, running_vars, cfg):
        parts = cfg["parts"]
        num_groups = cfg["num_groups"]
        eps = cfg["eps"]
        momentum = cfg["momentum"]
        training = cfg["training"]
        mode = cfg["mode"]
        relu = cfg["relu"]
        ns_iters = cfg.get("ns_iters", 7)
        track = cfg.get("track_running_stats", True)

        n, c, h, w = x.shape
        assert n % parts == 0, "batch not divisible by number of domain branches"
        b = n // parts
        g = c // num_groups

        comp_dtype = torch.float32 if x.dtype != torch.float64 else torch.float64
        y0 = torch.empty_like(x)
        means = []
        wmats = []
        saved_mat = []
        use_batch = training or not track
        stats_sync = cfg.get("stats_sync", False) and use_batch
        dist, world = _dist_world() if stats_sync else (None, 1)
        ctx.world = world
        for p in range(parts):
            xp = x[p * b:(p + 1) * b].to(comp_dtype)
            if use_batch:
                if dist is not None:
                    # cross-rank batch statistics (SyncBN-style): all-reduce
                    # the raw sums and co-moments, stats over the GLOBAL batch
                    m_local = xp.sum(dim=(0, 2, 3))
                    t = xp.permute(1, 0, 2, 3).reshape(num_groups, g, -1)
                    pm = torch.bmm(t, t.transpose(1, 2))
                    pack = torch.cat([m_local, pm.reshape(-1)])
                    dist.all_reduce(pack)
                    m_cnt = b * h * w * world
                    m = (pack[:c] / m_cnt).view(1, c, 1, 1)
                    exx = pack[c:].reshape(num_groups, g, g) / m_cnt
                    mg = m.reshape(num_groups, g, 1)
                    cov = exx - torch.bmm(mg, mg.transpose(1, 2))
                    xn = xp - m
                else:
                    m = xp.mean(dim=(0, 2, 3)).view(1, c, 1, 1)
                    xn = xp - m
                    cov = oracle.grouped_cov(xn, num_groups)
                cov_s = oracle.shrink_cov(cov, eps)
            else:
                m = running_means[p].reshape(1, c, 1, 1).to(comp_dtype)
                xn = xp - m
                cov = None
                cov_s = oracle.shrink_cov(running_vars[p].to(comp_dtype), eps)

            if mode == "chol":
                wmat, ell = matfn_chol_forward(cov_s)
                saved_mat.append(ell)
            else:
                wmat, saved = matfn_ns_forward(cov_s, ns_iters)
                saved_mat.append(saved)

            y0[p * b:(p + 1) * b] = _grouped_apply(xn, wmat, num_groups).to(x.dtype)
            means.append(m)
            wmats.append(wmat)

            if training and track and running_means is not None:
                with torch.no_grad():
                    rm = running_means[p]
                    rv = running_vars[p]
                    rm.mul_(1.0 - momentum).add_(m.reshape(rm.shape).to(rm.dtype), alpha=momentum)
                    rv.mul_(1.0 - momentum).add_(cov.reshape(rv.shape).to(rv.dtype), alpha=momentum)

        if gamma is not None:
            out = y0 * gamma.to(x.dtype) + beta.to(x.dtype)
        else:
            out = y0
        if relu:
            out = torch.relu(out)

        ctx.cfg = cfg
        ctx.comp_dtype = comp_dtype
        ctx.has_affine = gamma is not None
        ctx.saved_mat = saved_mat
        ctx.means = means
        ctx.wmats = wmats
        ctx.save_for_backward(x, gamma, y0, out)
        return out

    @staticmethod
    def backward(ctx, dout):
        cfg = ctx.cfg
        parts, num_groups, eps = cfg["parts"], cfg["num_groups"], cfg["eps"]
        mode, relu = cfg["mode"], cfg["relu"]
        training = cfg["training"]
        track = cfg.get("track_running_stats", True)
        use_batch = training or not track
        x, gamma, y0, out = ctx.saved_tensors
        comp_dtype = ctx.comp_dtype

        n, c, h, w = x.shape
        b = n // parts

        if relu:
            dy = dout * (out > 0).to(dout.dtype)
        else:
            dy = dout

        if ctx.has_affine:
            dgamma = (dy * y0).sum(dim=(0, 2, 3)).reshape(gamma.shape).to(gamma.dtype)
            dbeta = dy.sum(dim=(0, 2, 3)).reshape(gamma.shape).to(gamma.dtype)
            dy0 = (dy * gamma.to(dy.dtype)).to(comp_dtype)
        else:
            dgamma = dbeta = None
            dy0 = dy.to(comp_dtype)

        dx = torch.empty_like(x)
        world = getattr(ctx, "world", 1)
        stats_sync = cfg.get("stats_sync", False) and use_batch and world > 1
        dist = None
        if stats_sync:
            import torch.distributed as dist
        m_count = b * h * w * world
        for p in range(parts):
            sl = slice(p * b, (p + 1) * b)
            xp = x[sl].to(comp_dtype)
            xn = xp - ctx.means[p]
            dyp = dy0[sl]
            wmat = ctx.wmats[p]

            # direct path
            dt = _grouped_apply(dyp, wmat.transpose(-1, -2), num_groups)
            if use_batch:
                # through the whitening matrix and the covariance
                w_bar = _grouped_outer(dyp, xn, num_groups)
                if stats_sync:
                    dist.all_reduce(w_bar)
                if mode == "chol":
                    a_bar = matfn_chol_backward(wmat, ctx.saved_mat[p], w_bar)
                else:
                    a_bar = matfn_ns_backward(ctx.saved_mat[p], w_bar)
                s_mat = (1.0 - eps) / m_count * (a_bar + a_bar.transpose(-1, -2))
                dt = dt + _grouped_apply(xn, s_mat, num_groups)
                # through the mean (m = per-channel mean of ALL ranks' x)
                corr = dt.mean(dim=(0, 2, 3), keepdim=True)
                if stats_sync:
                    dist.all_reduce(corr)
                    corr /= world
                dt = dt - corr
            dx[sl] = dt.to(x.dtype)

        return dx, dgamma, dbeta, None, None, None


def whiten_multi(
    x: torch.Tensor,
    gamma: Optional[torch.Tensor],
    beta: Optional[torch.Tensor],
    running_means: Optional[Sequence[torch.Tensor]],
    running_vars: Optional[Sequence[torch.Tensor]],
    *,
    parts: int,
    num_groups: int,
    eps: float = 1e-3,
    momentum: float = 0.1,
    training: bool = True,
    mode: str = "chol",
    relu: bool = False,
    ns_iters: int = 7,
    track_running_stats: bool = True,
) -> torch.Tensor:
    cfg = dict(parts=parts, num_groups=num_groups, eps=eps, momentum=momentum,
               training=training, mode=mode, relu=relu, ns_iters=ns_iters,
               track_running_stats=track_running_stats)
    if os.environ.get("DWT_AMD_STATS_SYNC") == "1":
        # cross-rank batch statistics ('sync' stats mode):
        # on GPU the HIP kernels serve it with a partial-sums -> all-reduce
        # -> finalize split; torch path elsewhere
        cfg["stats_sync"] = True
    if x.is_cuda:
        from ..kernels import dispatch
        if dispatch.available():
            return dispatch.whiten_multi(x, gamma, beta, running_means, running_vars, cfg)
        dispatch.require_or_warn("whiten_multi")
    return WhitenMulti.apply(x, gamma, beta, running_means, running_vars, cfg)


# ----------------------------------------------------------------------------
# Fused multi-branch domain BatchNorm
# ----------------------------------------------------------------------------


class BatchNormMulti(torch.autograd.Function):
    """Per-domain BN over batch thirds/halves + shared affine + optional ReLU.

    Stock BN semantics per branch (batch_norm.py:54-69 / F.batch_norm):
    biased variance normalizes, unbiased variance goes into the EMA,
    running = (1-momentum) * running + momentum * batch.
    """

    @staticmethod
    def forward(ctx, x, gamma, beta, running_means, running_vars, cfg):
        parts = cfg["parts"]
        eps = cfg["eps"]
        momentum = cfg["momentum"]
        training = cfg["training"]
        relu = cfg["relu"]
        track = cfg.get("track_running_stats", True)
        use_batch = training or not track

        spatial = x.dim() == 4
        n = x.shape[0]
        c = x.shape[1]
        b = n // parts
        comp_dtype = torch.float32 if x.dtype != torch.float64 else torch.float64

        xhat = torch.empty_like(x)
        means, invstds = [], []
        red_dims = (0, 2, 3) if spatial else (0,)
        stats_sync = cfg.get("stats_sync", False) and use_batch
        dist, world = _dist_world() if stats_sync else (None, 1)
        ctx.world = world
        for p in range(parts):
            sl = slice(p * b, (p + 1) * b)
            xp = x[sl].to(comp_dtype)
            if use_batch:
                if dist is not None:
                    pack = torch.stack([xp.sum(dim=red_dims),
                                        (xp * xp).sum(dim=red_dims)])
                    dist.all_reduce(pack)
                    cnt = (xp.numel() // c) * world
                    m = pack[0] / cnt
                    var = (pack[1] / cnt - m * m).clamp_min(0)
                else:
                    m = xp.mean(dim=red_dims)
                    var = xp.var(dim=red_dims, unbiased=False)
                    cnt = xp.numel() // c
                if training and track and running_means is not None:
                    with torch.no_grad():
                        rm, rv = running_means[p], running_vars[p]
                        var_unb = var * (cnt / max(cnt - 1, 1))
                        rm.mul_(1 - momentum).add_(m.to(rm.dtype), alpha=momentum)
                        rv.mul_(1 - momentum).add_(var_unb.to(rv.dtype), alpha=momentum)
            else:
                m = running_means[p].to(comp_dtype)
                var = running_vars[p].to(comp_dtype)
            istd = torch.rsqrt(var + eps)
            shape = (1, c, 1, 1) if spatial else (1, c)
            xhat[sl] = ((xp - m.reshape(shape)) * istd.reshape(shape)).to(x.dtype)
            means.append(m)
            invstds.append(istd)

        if gamma is not None:
            out = xhat * gamma.to(x.dtype) + beta.to(x.dtype)
        else:
            out = xhat
        if relu:
            out = torch.relu(out)

        ctx.cfg = cfg
        ctx.means = means
        ctx.invstds = invstds
        ctx.spatial = spatial
        ctx.has_affine = gamma is not None
        ctx.comp_dtype = comp_dtype
        ctx.save_for_backward(x, gamma, xhat, out)
        return out

    @staticmethod
    def backward(ctx, dout):
        cfg = ctx.cfg
        parts, relu, training = cfg["parts"], cfg["relu"], cfg["training"]
        track = cfg.get("track_running_stats", True)
        use_batch = training or not track
        x, gamma, xhat, out = ctx.saved_tensors
        comp_dtype = ctx.comp_dtype
        spatial = ctx.spatial
        n, c = x.shape[0], x.shape[1]
        b = n // parts
        red_dims = (0, 2, 3) if spatial else (0,)
        shape = (1, c, 1, 1) if spatial else (1, c)

        dy = dout * (out > 0).to(dout.dtype) if relu else dout
        if ctx.has_affine:
            dgamma = (dy * xhat).sum(dim=red_dims).reshape(gamma.shape).to(gamma.dtype)
            dbeta = dy.sum(dim=red_dims).reshape(gamma.shape).to(gamma.dtype)
            dxhat_all = (dy * gamma.to(dy.dtype)).to(comp_dtype)
        else:
            dgamma = dbeta = None
            dxhat_all = dy.to(comp_dtype)

        dx = torch.empty_like(x)
        for p in range(parts):
            sl = slice(p * b, (p + 1) * b)
            dxh = dxhat_all[sl]
            istd = ctx.invstds[p].reshape(shape)
            xh = xhat[sl].to(comp_dtype)
            if use_batch:
                mean_dxh = dxh.mean(dim=red_dims, keepdim=True)
                mean_dxh_xh = (dxh * xh).mean(dim=red_dims, keepdim=True)
                world = getattr(ctx, "world", 1)
                if cfg.get("stats_sync", False) and world > 1:
                    import torch.distributed as dist
                    pack = torch.stack([mean_dxh, mean_dxh_xh])
                    dist.all_reduce(pack)
                    mean_dxh, mean_dxh_xh = pack[0] / world, pack[1] / world
                dxp = (dxh - mean_dxh - xh * mean_dxh_xh) * istd
            else:
                dxp = dxh * istd
            dx[sl] = dxp.to(x.dtype)
        return dx, dgamma, dbeta, None, None, None


def batch_norm_multi(
    x: torch.Tensor,
    gamma: Optional[torch.Tensor],
    beta: Optional[torch.Tensor],
    running_means: Optional[Sequence[torch.Tensor]],
    running_vars: Optional[Sequence[torch.Tensor]],
    *,
    parts: int,
    eps: float = 1e-5,
    momentum: float = 0.1,
    training: bool = True,
    relu: bool = False,
    track_running_stats: bool = True,
) -> torch.Tensor:
    cfg = dict(parts=parts, eps=eps, momentum=momentum, training=training,
               relu=relu, track_running_stats=track_running_stats)
    if os.environ.get("DWT_AMD_STATS_SYNC") == "1":
        cfg["stats_sync"] = True
    if x.is_cuda:
        from ..kernels import dispatch
        if dispatch.available():
            return dispatch.batch_norm_multi(x, gamma, beta, running_means, running_vars, cfg)
        dispatch.require_or_warn("batch_norm_multi")
    return BatchNormMulti.apply(x, gamma, beta, running_means, running_vars, cfg)


# ----------------------------------------------------------------------------
# Losses with explicit backwards
# ----------------------------------------------------------------------------


class MecLossFn(torch.autograd.Function):
    """mean_n min_k -1/2 (log p_x[k] + log p_y[k]); backward scatters into the
    per-sample argmin class: dx = (p_x - onehot(k*)) / (2N)."""

    @staticmethod
    def forward(ctx, x, y):
        ct = torch.float64 if x.dtype == torch.float64 else torch.float32
        lx = F.log_softmax(x.to(ct), dim=1)
        ly = F.log_softmax(y.to(ct), dim=1)
        per = -0.5 * (lx + ly)
        vals, idx = per.min(dim=1)
        ctx.save_for_backward(lx, ly, idx)
        ctx.dtypes = (x.dtype, y.dtype)
        return vals.mean()

    @staticmethod
    def backward(ctx, dloss):
        lx, ly, idx = ctx.saved_tensors
        n, k = lx.shape
        onehot = F.one_hot(idx, k).to(lx.dtype)
        dx = (lx.exp() - onehot) * (dloss.to(lx.dtype) / (2 * n))
        dy = (ly.exp() - onehot) * (dloss.to(ly.dtype) / (2 * n))
        return dx.to(ctx.dtypes[0]), dy.to(ctx.dtypes[1])


class EntropyLossFn(torch.autograd.Function):
    """H = mean_n -sum_k p log p ; dH/dx_j = -p_j (log p_j + H_n) / N."""

    @staticmethod
    def forward(ctx, x):
        ct = torch.float64 if x.dtype == torch.float64 else torch.float32
        q = F.log_softmax(x.to(ct), dim=1)
        p = q.exp()
        h_per = -(p * q).sum(dim=1)
        ctx.save_for_backward(p, q, h_per)
        ctx.dtype = x.dtype
        return h_per.mean()

    @staticmethod
    def backward(ctx, dloss):
        p, q, h_per = ctx.saved_tensors
        n = p.shape[0]
        dx = -p * (q + h_per.unsqueeze(1)) * (dloss.to(p.dtype) / n)
        return dx.to(ctx.dtype)


def mec_loss(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        from ..kernels import dispatch
        if dispatch.available():
            return dispatch.mec_loss(x, y)
        dispatch.require_or_warn("mec_loss")
    return MecLossFn.apply(x, y)


def entropy_loss(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        from ..kernels import dispatch
        if dispatch.available():
            return dispatch.entropy_loss(x)
        dispatch.require_or_warn("entropy_loss")
    return EntropyLossFn.apply(x)


# ----------------------------------------------------------------------------
# Fused residual join: relu(a + b)
# ----------------------------------------------------------------------------


class AddReluFn(torch.autograd.Function):
    """out = relu(a + b); da = db = dout * (out > 0) — one kernel each way on
    GPU (the reference's bottleneck exit, resnet50_dwt_mec_officehome.py:239)."""

    @staticmethod
    def _mf(t):
        if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last) \
                and not t.is_contiguous():
            return torch.channels_last
        return torch.contiguous_format

    @staticmethod
    def forward(ctx, a, b):
        if a.is_cuda and a.dtype in (torch.float32, torch.bfloat16):
            from ..kernels import dispatch
            if dispatch.available():
                ext = dispatch.ext()
                mf = AddReluFn._mf(a)
                ac = a.contiguous(memory_format=mf) if a.dim() == 4 else a.contiguous()
                bc = b.contiguous(memory_format=mf) if b.dim() == 4 else b.contiguous()
                out = torch.empty_like(ac)
                ext.add_relu_fwd(ac, bc, out)
                ctx.save_for_backward(out)
                ctx.use_hip = True
                return out
        out = torch.relu(a + b)
        ctx.save_for_backward(out)
        ctx.use_hip = False
        return out

    @staticmethod
    def backward(ctx, dout):
        (out,) = ctx.saved_tensors
        if ctx.use_hip:
            from ..kernels import dispatch
            ext = dispatch.ext()
            mf = AddReluFn._mf(out)
            dout = dout.contiguous(memory_format=mf) if dout.dim() == 4 \
                else dout.contiguous()
            din = torch.empty_like(out)
            ext.add_relu_bwd(dout, out, din)
            return din, din
        mask = (out > 0).to(dout.dtype)
        d = dout * mask
        return d, d


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return AddReluFn.apply(a, b)


def ce_loss(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Source cross-entropy: nll_loss(log_softmax(x)) (usps_mnist.py:298)."""
    if logits.is_cuda:
        from ..kernels import dispatch
        if dispatch.available():
            return dispatch.ce_loss(logits, target)
        dispatch.require_or_warn("ce_loss")
    return F.nll_loss(F.log_softmax(logits.float(), dim=1), target)
