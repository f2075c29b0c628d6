"""Python layer over the hand-written MFMA GEMM / implicit-GEMM conv kernels.

``mfma_gemm``: C[M,N] = A[M,K] @ B[N,K]^T (+bias/+ReLU), bf16 in, fp32
accumulate, bf16 out — the v_mfma_f32_16x16x32_bf16 tile kernel.

``MFMALinear``: nn.Linear whose forward AND backward run on mfma_gemm
(dgrad/wgrad are plain GEMMs with small host-side transposes).

``conv2d_fwd``: NHWC implicit-GEMM convolution forward on the same kernel
(1x1 degenerates to the dense-GEMM fast path; KxK gathers patches on the
fly; the channels_last conv weight is consumed as stored).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _ext():
    from ..kernels import dispatch
    return dispatch.ext()


def _pad_k(t: torch.Tensor) -> torch.Tensor:
    k = t.shape[1]
    if k % 8 == 0:
        return t
    return F.pad(t, (0, 8 - k % 8))


def mfma_gemm(a: torch.Tensor, bt: torch.Tensor,
              bias: Optional[torch.Tensor] = None, relu: bool = False):
    """a (M,K) bf16, bt (N,K) bf16 -> (M,N) bf16 = a @ bt.T."""
    assert a.dtype == torch.bfloat16 and bt.dtype == torch.bfloat16
    a = _pad_k(a.contiguous())
    bt = _pad_k(bt.contiguous())
    m, n = a.shape[0], bt.shape[0]
    out = torch.empty(m, n, device=a.device, dtype=torch.bfloat16)
    has_bias = bias is not None
    b32 = bias.detach().float().contiguous() if has_bias else torch.empty(0, device=a.device)
    _ext().mfma_gemm(a, bt, b32, out, relu, has_bias)
    return out


class _MFMALinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        x2 = x.reshape(-1, x.shape[-1])
        out = mfma_gemm(x2, weight, bias=bias, relu=False)
        ctx.save_for_backward(x2, weight)
        ctx.has_bias = bias is not None
        ctx.in_shape = x.shape
        return out.reshape(*x.shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dout):
        x2, weight = ctx.saved_tensors
        dout2 = dout.reshape(-1, dout.shape[-1]).contiguous()
        # dx = dy @ W          : A = dy (M,N), Bt = W^T (K,N)
        dx = mfma_gemm(dout2, weight.t().contiguous())
        # dW = dy^T @ x        : A = dy^T (N,M), Bt = x^T (K,M)
        dw = mfma_gemm(dout2.t().contiguous(), x2.t().contiguous())
        db = dout2.sum(0).to(dout.dtype) if ctx.has_bias else None
        return dx.reshape(ctx.in_shape), dw, db


class MFMALinear(nn.Linear):
    """nn.Linear running fwd/dgrad/wgrad on the hand-written MFMA GEMM when
    on GPU in bf16; falls back to F.linear otherwise."""

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and self.weight.dtype == torch.bfloat16:
            from ..kernels import dispatch
            if dispatch.available():
                return _MFMALinearFn.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)


def conv2d_fwd(x: torch.Tensor, weight: torch.Tensor,
               bias: Optional[torch.Tensor] = None, stride: int = 1,
               padding: int = 0, relu: bool = False) -> torch.Tensor:
    """NHWC implicit-GEMM conv forward (inference path).

    x: (N, C, H, W) channels_last bf16; weight: (Cout, Cin, KH, KW)
    channels_last bf16; returns (N, Cout, P, Q) channels_last bf16.
    """
    assert x.dtype == torch.bfloat16
    assert x.is_contiguous(memory_format=torch.channels_last) or x.shape[1] == 1
    n, cin, h, w = x.shape
    cout, _, kh, kw = weight.shape
    if cin == 3:
        # stem: zero-pad to 4 channels so the kernel's paired-pixel 16-B
        # path applies (NHWC C=4 chunks); weight padded to match
        x4 = torch.empty(n, 4, h, w, device=x.device, dtype=x.dtype,
                         memory_format=torch.channels_last)
        x4[:, 3] = 0
        x4[:, :3] = x
        w4 = torch.empty(cout, 4, kh, kw, device=x.device, dtype=weight.dtype,
                         memory_format=torch.channels_last)
        w4[:, 3] = 0
        w4[:, :3] = weight.detach()
        x, weight, cin = x4, w4, 4
    p = (h + 2 * padding - kh) // stride + 1
    q = (w + 2 * padding - kw) // stride + 1
    out = torch.empty(n, cout, p, q, device=x.device, dtype=torch.bfloat16,
                      memory_format=torch.channels_last)
    wgt = weight
    if not wgt.is_contiguous(memory_format=torch.channels_last):
        wgt = wgt.contiguous(memory_format=torch.channels_last)
    has_bias = bias is not None
    b32 = bias.detach().float().contiguous() if has_bias else torch.empty(0, device=x.device)
    if (kh * kw * cin) % 8 != 0 and not (kh == 1 and kw == 1 and stride == 1 and padding == 0):
        pass  # slow per-element gather path inside the kernel handles it
    _ext().mfma_conv2d_fwd(x, wgt, b32, out, n, h, w, cin, p, q, cout,
                           kh, kw, stride, padding, relu, has_bias)
    return out


def conv2d_dgrad(dy: torch.Tensor, weight: torch.Tensor, in_shape,
                 stride: int = 1, padding: int = 0) -> torch.Tensor:
    """dx for an NHWC conv: implicit GEMM over dy with the weight permuted to
    [ci][r][s][co] (tiny host-side permute per call)."""
    n, cin, h, w = in_shape
    cout, _, kh, kw = weight.shape
    p, q = dy.shape[2], dy.shape[3]
    dyc = dy.contiguous(memory_format=torch.channels_last)
    # CL storage of weight is [co][kh][kw][ci]; build [ci][kh][kw][co]
    wv = weight.permute(0, 2, 3, 1)              # (Cout, KH, KW, Cin), contig view of CL
    wd = wv.permute(3, 1, 2, 0).contiguous()     # (Cin, KH, KW, Cout)
    dx = torch.empty(n, cin, h, w, device=dy.device, dtype=torch.bfloat16,
                     memory_format=torch.channels_last)
    _ext().mfma_conv2d_dgrad(dyc, wd, dx, n, h, w, cin, p, q, cout,
                             kh, kw, stride, padding)
    return dx


class _MFMAConv2dFn(torch.autograd.Function):
    """Training conv on the hand-written MFMA kernels: implicit-GEMM fwd,
    dgrad (skipped when the input needs no grad, e.g. the stem) and wgrad.
    DWT_AMD_WGRAD=aten routes wgrad through the library as an escape hatch."""

    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        out = conv2d_fwd(x, weight, bias=bias, stride=stride, padding=padding)
        ctx.save_for_backward(x, weight)
        ctx.meta = (stride, padding, bias is not None)
        return out

    @staticmethod
    def backward(ctx, dout):
        import os
        x, weight = ctx.saved_tensors
        stride, padding, has_bias = ctx.meta
        dout = dout.contiguous(memory_format=torch.channels_last)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = conv2d_dgrad(dout, weight, x.shape, stride=stride,
                              padding=padding)
        if os.environ.get("DWT_AMD_WGRAD") == "aten":
            _, dw, db = torch.ops.aten.convolution_backward(
                dout, x, weight, [weight.shape[0]] if has_bias else None,
                [stride, stride], [padding, padding], [1, 1], False, [0, 0], 1,
                [False, True, has_bias])
        else:
            dw = conv2d_wgrad(dout, x, tuple(weight.shape), stride=stride,
                              padding=padding)
            db = dout.sum(dim=(0, 2, 3)).to(dout.dtype) if has_bias else None
        return dx, dw, db, None, None


class MFMAConv2d(nn.Conv2d):
    """Conv2d on the hand-written MFMA kernels (fwd + dgrad) when running
    bf16 channels_last on GPU; F.conv2d otherwise.  Enable in the models with
    DWT_AMD_CONV=hip."""

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and (x.shape[1] >= 8 or x.shape[1] == 3) \
                and x.is_contiguous(memory_format=torch.channels_last) \
                and self.stride[0] == self.stride[1] \
                and self.padding[0] == self.padding[1]:
            from ..kernels import dispatch
            if dispatch.available():
                return _MFMAConv2dFn.apply(x, self.weight, self.bias,
                                           self.stride[0], self.padding[0])
        return super().forward(x)


def _wgrad_patches_reference(x_nhwc, shape, stride, padding, npq_pad=None,
                             dtype=torch.float64):
    """CPU/torch reference of the kernel's wgrad B-operand gather: returns
    Bt [(r,s,ci), k=npq] built with EXACTLY the kernel's index decode
    (mfma_conv.hip::load_b_wgrad) — used by the index-math test."""
    n, h, w, cin = x_nhwc.shape
    cout, _, kh, kw = shape
    p = (h + 2 * padding - kh) // stride + 1
    q = (w + 2 * padding - kw) // stride + 1
    npq = n * p * q
    k_tot = npq_pad or npq
    cols = torch.zeros(kh * kw * cin, k_tot, dtype=dtype)
    ks = torch.arange(npq)
    qq = ks % q
    pp = (ks // q) % p
    nn = ks // (q * p)
    for r in range(kh):
        for s in range(kw):
            hh = pp * stride - padding + r
            ww = qq * stride - padding + s
            ok = (hh >= 0) & (hh < h) & (ww >= 0) & (ww < w)
            vals = torch.zeros(npq, cin, dtype=dtype)
            vals[ok] = x_nhwc[nn[ok], hh[ok].clamp(0), ww[ok].clamp(0)].to(dtype)
            for ci in range(cin):
                cols[(r * kw + s) * cin + ci, :npq] = vals[:, ci]
    return cols


def conv2d_wgrad(dy: torch.Tensor, x: torch.Tensor, weight_shape,
                 stride: int = 1, padding: int = 0) -> torch.Tensor:
    """dw for an NHWC conv on the split-K MFMA wgrad kernel: both operands
    consumed K-major straight from their NHWC storage (LDS-transposed in
    kernel), K = N*P*Q split across the grid into an fp32 workspace;
    output IS the channels_last weight storage [co][r][s][ci]."""
    cout, cin, kh, kw = weight_shape
    n, _, p, q = dy.shape
    assert cout % 8 == 0, "wgrad kernel needs Cout % 8 == 0"
    dyc = dy.contiguous(memory_format=torch.channels_last)
    dw = torch.empty(weight_shape, device=dy.device, dtype=torch.bfloat16,
                     memory_format=torch.channels_last)
    ws = torch.zeros(cout * kh * kw * cin, device=dy.device,
                     dtype=torch.float32)
    xc = x.contiguous(memory_format=torch.channels_last)
    _ext().mfma_conv2d_wgrad(dyc, xc, dw, ws, n, x.shape[2], x.shape[3], cin,
                             p, q, cout, kh, kw, stride, padding)
    return dw
