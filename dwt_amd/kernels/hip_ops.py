"""GPU autograd Functions backed by the gfx950 HIP kernels.

Mirrors the algorithm contract of dwt_amd/ops/functional.py exactly (these
are numerically tested against it in tests/test_gpu_kernels.py).  The
register-blocked fast path covers group sizes {2, 4, 8}; 8 < g <= 32 runs
the generic LDS-tiled NCHW kernels (every group size the reference models
can construct); fp32/bf16 inputs; anything else falls back to the torch
implementation (warned once).
"""
from __future__ import annotations

import warnings
import torch

from . import dispatch

_GEN_G_MAX = 32  # register-blocked kernels cover g in {2,4,8};
                 # 8 < g <= 32 runs the LDS-tiled generic NCHW kernels
_warned = set()


def _whiten_layout(x, c, g):
    """'cl' when x is channels_last and the NHWC kernels support (C, g):
    the lane map needs GW = min(C,256)/g to divide the 256-thread block.
    g > 8 always routes NCHW (the generic LDS-tiled kernels)."""
    if x.dim() == 4 and x.is_contiguous(memory_format=torch.channels_last) \
            and not x.is_contiguous():
        if c % 64 == 0 and (c <= 256 or c % 256 == 0) and g in (2, 4):
            # g=8 NHWC instantiations spill registers (profiles/
            # kernel_resources.md) — route them to the NCHW path
            gw = min(c, 256) // g
            if gw > 0 and 256 % gw == 0:
                return "cl"
    return "nchw"


def _bn_cl_supported(c):
    """The NHWC BN kernels map lanes as t %% NCH with NCH = min(C,1024)/4;
    rows advance by blockDim/NCH — NCH must divide the block (else boundary
    rows are double-counted)."""
    if c % 4 != 0 or (c > 1024 and c % 1024 != 0):
        return False
    nch = min(c, 1024) // 4
    return 256 % nch == 0


def _bn_layout(x, c):
    if x.dim() == 2 and x.is_contiguous() and _bn_cl_supported(c):
        return "cl"  # (N, C) IS channels-last: C contiguous per position
    if x.dim() == 4 and x.is_contiguous(memory_format=torch.channels_last) \
            and not x.is_contiguous() and _bn_cl_supported(c):
        return "cl"
    return "nchw"


def _warn_once(key, msg):
    if key not in _warned:
        _warned.add(key)
        warnings.warn(msg)


def _sync_world():
    """World size for cross-rank ('sync') batch statistics."""
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized():
        return dist, dist.get_world_size()
    return None, 1


def _allreduce(t):
    import torch.distributed as dist
    dist.all_reduce(t)
    return t


def _ext():
    return dispatch.ext()


def _f32c(t):
    return t.detach().to(torch.float32).contiguous()


class _HipWhitenMulti(torch.autograd.Function):
    """All per-branch statistics are STACKED ([parts*C] means, [parts*G, g, g]
    covariances/W) so the channels_last path runs ONE launch per pass across
    all domain branches (grid.y = branch — late-layer sites are launch-bound
    otherwise, profiles/norm_bench.md); the matrix-function kernels always
    see the stacked group axis.  The NCHW path loops branches over contiguous
    slices of the same stacked tensors."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_means, running_vars, cfg):
        ext = _ext()
        parts = cfg["parts"]
        g = x.shape[1] // cfg["num_groups"]
        eps, momentum = cfg["eps"], cfg["momentum"]
        training, mode, relu = cfg["training"], cfg["mode"], cfg["relu"]
        track = cfg.get("track_running_stats", True)
        ns_iters = cfg.get("ns_iters", 7)
        use_batch = training or not track

        c = x.shape[1]
        layout = _whiten_layout(x, c, g)
        if layout == "nchw":
            x = x.contiguous()
        n, c, h, w = x.shape
        assert n % parts == 0,             f"batch {n} not divisible by {parts} domain branches"
        b = n // parts
        n_groups = c // g
        ng_all = parts * n_groups
        dev = x.device
        m_count = b * h * w

        has_affine = gamma is not None
        gflat = gamma.detach().reshape(c).contiguous() if has_affine else torch.empty(0, device=dev, dtype=x.dtype)
        bflat = beta.detach().reshape(c).contiguous() if has_affine else torch.empty(0, device=dev, dtype=x.dtype)

        stats_sync = cfg.get("stats_sync", False) and use_batch
        world = 1
        if stats_sync:
            _, world = _sync_world()
        ctx_count = m_count * world  # global positions per branch

        out = torch.empty_like(x)
        if use_batch:
            acc = torch.zeros(ng_all * (g + g * g), device=dev, dtype=torch.float32)
            mean = torch.empty(parts * c, device=dev, dtype=torch.float32)
            cov = torch.empty(ng_all, g, g, device=dev, dtype=torch.float32)
            if stats_sync:
                # split pass: local raw sums -> all-reduce -> finalize over
                # the GLOBAL batch (SyncBN-style cross-rank statistics)
                if layout == "cl":
                    ext.whiten_stats_partial_cl(x, acc, g, c, m_count, parts)
                else:
                    for p in range(parts):
                        ext.whiten_stats_partial(
                            x[p * b:(p + 1) * b],
                            acc[p * n_groups * (g + g * g):(p + 1) * n_groups * (g + g * g)], g)
                if world > 1:
                    _allreduce(acc)
                ext.whiten_stats_final(acc, mean, cov, g, ng_all, ctx_count)
            elif layout == "cl":
                ext.whiten_stats_cl(x, acc, mean, cov, g, c, m_count, parts)
            else:
                for p in range(parts):
                    ext.whiten_stats(x[p * b:(p + 1) * b],
                                     acc[p * n_groups * (g + g * g):(p + 1) * n_groups * (g + g * g)],
                                     mean[p * c:(p + 1) * c],
                                     cov[p * n_groups:(p + 1) * n_groups], g)
        else:
            mean = torch.stack([_f32c(running_means[p]).reshape(c)
                                for p in range(parts)]).reshape(-1).contiguous()
            cov = torch.stack([_f32c(running_vars[p]).reshape(n_groups, g, g)
                               for p in range(parts)]).reshape(ng_all, g, g).contiguous()

        wmat = torch.empty(ng_all, g, g, device=dev, dtype=torch.float32)
        if mode == "chol":
            ell = torch.empty_like(wmat)
            ext.matfn_chol_fwd(cov, wmat, ell, eps)
            saved_mat = ell
        else:
            ys = torch.empty(ng_all, ns_iters, g, g, device=dev, dtype=torch.float32)
            zs = torch.empty_like(ys)
            svals = torch.empty(ng_all, device=dev, dtype=torch.float32)
            ext.matfn_ns_fwd(cov, wmat, ys, zs, svals, eps, ns_iters)
            saved_mat = (ys, zs, svals)

        if layout == "cl":
            ext.whiten_apply_cl(x, mean, wmat, gflat, bflat, out, g, c,
                                m_count, relu, has_affine, parts)
        else:
            for p in range(parts):
                ext.whiten_apply(x[p * b:(p + 1) * b], mean[p * c:(p + 1) * c],
                                 wmat[p * n_groups:(p + 1) * n_groups], gflat,
                                 bflat, out[p * b:(p + 1) * b], g, relu,
                                 has_affine)

        if training and track and running_means is not None:
            with torch.no_grad():
                for p in range(parts):
                    rm, rv = running_means[p], running_vars[p]
                    mslice = mean[p * c:(p + 1) * c]
                    cslice = cov[p * n_groups:(p + 1) * n_groups]
                    if rm.dtype == torch.float32 and rm.is_contiguous() \
                            and rv.dtype == torch.float32 and rv.is_contiguous():
                        ext.ema_update(rm, mslice, rv, cslice, momentum)
                    else:
                        rm.mul_(1.0 - momentum).add_(
                            mslice.reshape(rm.shape).to(rm.dtype), alpha=momentum)
                        rv.mul_(1.0 - momentum).add_(
                            cslice.reshape(rv.shape).to(rv.dtype), alpha=momentum)

        ctx.cfg = cfg
        ctx.g = g
        ctx.layout = layout
        ctx.m_count = ctx_count
        ctx.sync_world = world
        ctx.mean = mean
        ctx.wmat = wmat
        ctx.saved_mat = saved_mat
        ctx.has_affine = has_affine
        ctx.gflat = gflat
        ctx.bflat = bflat
        ctx.save_for_backward(x, gamma, out)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _ext()
        cfg = ctx.cfg
        parts, relu, mode, eps = cfg["parts"], cfg["relu"], cfg["mode"], cfg["eps"]
        training = cfg["training"]
        track = cfg.get("track_running_stats", True)
        ns_iters = cfg.get("ns_iters", 7)
        use_batch = training or not track
        g = ctx.g
        x, gamma, out = ctx.saved_tensors
        n, c, h, w = x.shape
        b = n // parts
        n_groups = c // g
        ng_all = parts * n_groups
        dev = x.device
        layout = ctx.layout
        if layout == "cl":
            dout = dout.contiguous(memory_format=torch.channels_last)
        else:
            dout = dout.contiguous()
        mean, wmat = ctx.mean, ctx.wmat
        m_count = ctx.m_count

        dgb = torch.zeros(parts, 2, c, device=dev, dtype=torch.float32)
        dW = torch.zeros(ng_all, g, g, device=dev, dtype=torch.float32)
        dx = torch.empty_like(x)

        if layout == "cl":
            ext.whiten_bwd_reduce_cl(x, dout, mean, wmat, ctx.gflat, ctx.bflat,
                                     dW, dgb.reshape(-1), g, c, m_count,
                                     relu, ctx.has_affine, parts)
        else:
            for p in range(parts):
                sl = slice(p * b, (p + 1) * b)
                ext.whiten_bwd_reduce(x[sl], dout[sl], out[sl],
                                      mean[p * c:(p + 1) * c],
                                      wmat[p * n_groups:(p + 1) * n_groups],
                                      ctx.gflat, dW[p * n_groups:(p + 1) * n_groups],
                                      dgb[p].reshape(-1), g, relu, ctx.has_affine)

        if use_batch:
            gdb = (ctx.gflat.float().unsqueeze(0) * dgb[:, 1]) if ctx.has_affine \
                else dgb[:, 1]
            gdb = gdb.reshape(-1).contiguous()
            dW_in = dW
            if getattr(ctx, "sync_world", 1) > 1:
                # global-batch statistics: the matrix-function backward sees
                # the cross-rank sums (dgamma/dbeta stay LOCAL — the DP
                # gradient all-reduce handles parameter grads)
                dW_in = _allreduce(dW.clone())
                gdb = _allreduce(gdb.clone())
            S = torch.empty(ng_all, g, g, device=dev, dtype=torch.float32)
            corr = torch.empty(parts * c, device=dev, dtype=torch.float32)
            inv_m = 1.0 / m_count
            if mode == "chol":
                ext.matfn_chol_bwd(dW_in, wmat, ctx.saved_mat, gdb, S, corr,
                                   eps, inv_m)
            else:
                ys, zs, svals = ctx.saved_mat
                ext.matfn_ns_bwd(dW_in, wmat, ys, zs, svals, gdb, S, corr,
                                 eps, inv_m, ns_iters)
        else:
            S = torch.empty(0, device=dev)
            corr = torch.empty(0, device=dev)

        if layout == "cl":
            ext.whiten_bwd_apply_cl(x, dout, mean, wmat, ctx.gflat, ctx.bflat,
                                    S, corr, dx, g, c, m_count, relu,
                                    ctx.has_affine, use_batch, parts)
        else:
            for p in range(parts):
                sl = slice(p * b, (p + 1) * b)
                Sp = S[p * n_groups:(p + 1) * n_groups] if use_batch else S
                cp = corr[p * c:(p + 1) * c] if use_batch else corr
                ext.whiten_bwd_apply(x[sl], dout[sl], out[sl],
                                     mean[p * c:(p + 1) * c],
                                     wmat[p * n_groups:(p + 1) * n_groups],
                                     ctx.gflat, Sp, cp, dx[sl], g, relu,
                                     ctx.has_affine, use_batch)

        if ctx.has_affine:
            dgamma = dgb[:, 0].sum(0).reshape(gamma.shape).to(gamma.dtype)
            dbeta = dgb[:, 1].sum(0).reshape(gamma.shape).to(gamma.dtype)
        else:
            dgamma = dbeta = None
        return dx, dgamma, dbeta, None, None, None


def whiten_multi(x, gamma, beta, running_means, running_vars, cfg):
    g = x.shape[1] // cfg["num_groups"]
    if g > _GEN_G_MAX or x.dtype not in (torch.float32, torch.bfloat16):
        _warn_once(("wh", g, x.dtype),
                   f"dwt_amd: whiten_multi g={g} dtype={x.dtype} uses the "
                   "torch path (HIP kernels cover g <= 32, fp32/bf16)")
        from ..ops.functional import WhitenMulti
        return WhitenMulti.apply(x, gamma, beta, running_means, running_vars, cfg)
    return _HipWhitenMulti.apply(x, gamma, beta, running_means, running_vars, cfg)


class _HipBatchNormMulti(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_means, running_vars, cfg):
        ext = _ext()
        parts = cfg["parts"]
        eps, momentum = cfg["eps"], cfg["momentum"]
        training, relu = cfg["training"], cfg["relu"]
        track = cfg.get("track_running_stats", True)
        use_batch = training or not track

        spatial = x.dim() == 4
        c = x.shape[1]
        layout = _bn_layout(x, c)
        if layout == "nchw" or x.dim() == 2:
            x = x.contiguous()
        n = x.shape[0]
        assert n % parts == 0,             f"batch {n} not divisible by {parts} domain branches"
        b = n // parts
        dev = x.device
        cnt = (x.numel() // parts) // c

        has_affine = gamma is not None
        gflat = gamma.detach().reshape(c).contiguous() if has_affine else torch.empty(0, device=dev, dtype=x.dtype)
        bflat = beta.detach().reshape(c).contiguous() if has_affine else torch.empty(0, device=dev, dtype=x.dtype)

        stats_sync = cfg.get("stats_sync", False) and use_batch
        world = 1
        if stats_sync:
            _, world = _sync_world()
        cnt_total = cnt * world

        out = torch.empty_like(x)
        if use_batch:
            acc = torch.zeros(parts * 2 * c, device=dev, dtype=torch.float32)
            mean = torch.empty(parts * c, device=dev, dtype=torch.float32)
            istd = torch.empty(parts * c, device=dev, dtype=torch.float32)
            var_unb = torch.empty(parts * c, device=dev, dtype=torch.float32)
            if stats_sync:
                if layout == "cl":
                    ext.bn_stats_partial_cl(x, acc, c, cnt, parts)
                else:
                    for p in range(parts):
                        ext.bn_stats_partial(x[p * b:(p + 1) * b],
                                             acc[p * 2 * c:(p + 1) * 2 * c])
                if world > 1:
                    _allreduce(acc)
                ext.bn_stats_final(acc, mean, istd, var_unb, c, parts,
                                   cnt_total, eps)
            elif layout == "cl":
                ext.bn_stats_cl(x, acc, mean, istd, var_unb, c, cnt, eps, parts)
            else:
                for p in range(parts):
                    ext.bn_stats(x[p * b:(p + 1) * b],
                                 acc[p * 2 * c:(p + 1) * 2 * c],
                                 mean[p * c:(p + 1) * c],
                                 istd[p * c:(p + 1) * c],
                                 var_unb[p * c:(p + 1) * c], eps)
            if training and track and running_means is not None:
                with torch.no_grad():
                    for p in range(parts):
                        rm, rv = running_means[p], running_vars[p]
                        mslice = mean[p * c:(p + 1) * c]
                        vslice = var_unb[p * c:(p + 1) * c]
                        if rm.dtype == torch.float32 and rm.is_contiguous() \
                                and rv.dtype == torch.float32 and rv.is_contiguous():
                            ext.ema_update(rm, mslice, rv, vslice, momentum)
                        else:
                            rm.mul_(1 - momentum).add_(mslice.to(rm.dtype), alpha=momentum)
                            rv.mul_(1 - momentum).add_(vslice.to(rv.dtype), alpha=momentum)
        else:
            mean = torch.stack([_f32c(running_means[p]).reshape(c)
                                for p in range(parts)]).reshape(-1).contiguous()
            var = torch.stack([_f32c(running_vars[p]).reshape(c)
                               for p in range(parts)]).reshape(-1)
            istd = torch.rsqrt(var + eps).contiguous()

        if layout == "cl":
            ext.bn_apply_cl(x, mean, istd, gflat, bflat, out, c, cnt, relu,
                            has_affine, parts)
        else:
            for p in range(parts):
                ext.bn_apply(x[p * b:(p + 1) * b], mean[p * c:(p + 1) * c],
                             istd[p * c:(p + 1) * c], gflat, bflat,
                             out[p * b:(p + 1) * b], relu, has_affine)

        ctx.cfg = cfg
        ctx.layout = layout
        ctx.cnt = cnt
        ctx.cnt_total = cnt_total
        ctx.sync_world = world
        ctx.spatial = spatial
        ctx.mean = mean
        ctx.istd = istd
        ctx.has_affine = has_affine
        ctx.gflat = gflat
        ctx.bflat = bflat
        ctx.use_batch = use_batch
        ctx.save_for_backward(x, gamma, out)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _ext()
        cfg = ctx.cfg
        parts, relu = cfg["parts"], cfg["relu"]
        x, gamma, out = ctx.saved_tensors
        n, c = x.shape[0], x.shape[1]
        b = n // parts
        dev = x.device
        layout = ctx.layout
        if layout == "cl" and x.dim() == 4:
            dout = dout.contiguous(memory_format=torch.channels_last)
        else:
            dout = dout.contiguous()
        mean, istd = ctx.mean, ctx.istd

        sums = torch.zeros(parts, 2, c, device=dev, dtype=torch.float32)
        dx = torch.empty_like(x)
        cnt_total = getattr(ctx, "cnt_total", ctx.cnt)
        if layout == "cl":
            ext.bn_bwd_reduce_cl(x, dout, mean, istd, ctx.gflat, ctx.bflat,
                                 sums.reshape(-1), c, ctx.cnt, relu,
                                 ctx.has_affine, parts)
            apply_sums = sums
            if getattr(ctx, "sync_world", 1) > 1:
                apply_sums = _allreduce(sums.clone())
            ext.bn_bwd_apply_cl(x, dout, mean, istd, ctx.gflat, ctx.bflat,
                                apply_sums.reshape(-1), dx, c, ctx.cnt, relu,
                                ctx.has_affine, ctx.use_batch, parts,
                                cnt_total)
        else:
            for p in range(parts):
                sl = slice(p * b, (p + 1) * b)
                ext.bn_bwd_reduce(x[sl], dout[sl], out[sl],
                                  mean[p * c:(p + 1) * c],
                                  istd[p * c:(p + 1) * c],
                                  sums[p].reshape(-1), relu)
            apply_sums = sums
            if getattr(ctx, "sync_world", 1) > 1:
                apply_sums = _allreduce(sums.clone())
            for p in range(parts):
                sl = slice(p * b, (p + 1) * b)
                ext.bn_bwd_apply(x[sl], dout[sl], out[sl],
                                 mean[p * c:(p + 1) * c],
                                 istd[p * c:(p + 1) * c],
                                 ctx.gflat, apply_sums[p].reshape(-1), dx[sl],
                                 relu, ctx.has_affine, ctx.use_batch,
                                 cnt_total)

        if ctx.has_affine:
            # dgamma = sum dy * xhat ; dbeta = sum dy  (sums[:,1] is dy*xhat)
            dgamma = sums[:, 1].sum(0).reshape(gamma.shape).to(gamma.dtype)
            dbeta = sums[:, 0].sum(0).reshape(gamma.shape).to(gamma.dtype)
        else:
            dgamma = dbeta = None
        return dx, dgamma, dbeta, None, None, None


def batch_norm_multi(x, gamma, beta, running_means, running_vars, cfg):
    if x.dtype not in (torch.float32, torch.bfloat16):
        from ..ops.functional import BatchNormMulti
        return BatchNormMulti.apply(x, gamma, beta, running_means, running_vars, cfg)
    return _HipBatchNormMulti.apply(x, gamma, beta, running_means, running_vars, cfg)


class _HipMecLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, y):
        ext = _ext()
        x32 = x.detach().to(torch.float32).contiguous()
        y32 = y.detach().to(torch.float32).contiguous()
        n, k = x32.shape
        lx = torch.empty_like(x32)
        ly = torch.empty_like(y32)
        amin = torch.empty(n, device=x.device, dtype=torch.int32)
        loss = torch.zeros(1, device=x.device, dtype=torch.float32)
        ext.mec_fwd(x32, y32, lx, ly, amin, loss)
        ctx.save_for_backward(lx, ly, amin)
        ctx.dtypes = (x.dtype, y.dtype)
        return loss[0]

    @staticmethod
    def backward(ctx, dloss):
        ext = _ext()
        lx, ly, amin = ctx.saved_tensors
        gscale = dloss.detach().to(torch.float32).reshape(1).contiguous()
        dx = torch.empty_like(lx)
        dy = torch.empty_like(ly)
        ext.mec_bwd(lx, ly, amin, gscale, dx, dy)
        return dx.to(ctx.dtypes[0]), dy.to(ctx.dtypes[1])


class _HipEntropyLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = _ext()
        x32 = x.detach().to(torch.float32).contiguous()
        n, k = x32.shape
        q = torch.empty_like(x32)
        hper = torch.empty(n, device=x.device, dtype=torch.float32)
        loss = torch.zeros(1, device=x.device, dtype=torch.float32)
        ext.entropy_fwd(x32, q, hper, loss)
        ctx.save_for_backward(q, hper)
        ctx.dtype = x.dtype
        return loss[0]

    @staticmethod
    def backward(ctx, dloss):
        ext = _ext()
        q, hper = ctx.saved_tensors
        gscale = dloss.detach().to(torch.float32).reshape(1).contiguous()
        dx = torch.empty_like(q)
        ext.entropy_bwd(q, hper, gscale, dx)
        return dx.to(ctx.dtype)


def mec_loss(x, y):
    return _HipMecLoss.apply(x, y)


def entropy_loss(x):
    return _HipEntropyLoss.apply(x)


class _HipCELoss(torch.autograd.Function):
    """mean_n -log softmax(x)[n, target_n] (reference source loss)."""

    @staticmethod
    def forward(ctx, x, target):
        ext = _ext()
        x32 = x.detach().to(torch.float32).contiguous()
        q = torch.empty_like(x32)
        loss = torch.zeros(1, device=x.device, dtype=torch.float32)
        ext.ce_fwd(x32, target.contiguous(), q, loss)
        ctx.save_for_backward(q, target)
        ctx.dtype = x.dtype
        return loss[0]

    @staticmethod
    def backward(ctx, dloss):
        ext = _ext()
        q, target = ctx.saved_tensors
        gscale = dloss.detach().to(torch.float32).reshape(1).contiguous()
        dx = torch.empty_like(q)
        ext.ce_bwd(q, target.contiguous(), gscale, dx)
        return dx.to(ctx.dtype), None


def ce_loss(x, target):
    return _HipCELoss.apply(x, target)
