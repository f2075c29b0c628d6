"""Pooling ops: NHWC maxpool fwd/bwd (argmax-index forward,
gather backward — no atomics) and global average pool, with transparent
fallback to torch pooling off-GPU / off-channels_last."""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F


def _cl4(t):
    return t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last) \
        and not t.is_contiguous()


def _hip_ok(x):
    if os.environ.get("DWT_AMD_POOL") == "torch":
        return False
    if not x.is_cuda or x.dtype not in (torch.float32, torch.bfloat16):
        return False
    from ..kernels import dispatch
    if not dispatch.available():
        dispatch.require_or_warn("pooling")  # no silent eager fallback on GPU
        return False
    return True


class MaxPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ks, stride, pad):
        from ..kernels import dispatch
        ext = dispatch.ext()
        n, c, h, w = x.shape
        p = (h + 2 * pad - ks) // stride + 1
        q = (w + 2 * pad - ks) // stride + 1
        out = torch.empty(n, c, p, q, device=x.device, dtype=x.dtype,
                          memory_format=torch.channels_last)
        idx = torch.empty(n * p * q * c, device=x.device, dtype=torch.uint8)
        ext.maxpool_cl_fwd(x, out, idx, c, h, w, p, q, ks, stride, pad)
        ctx.save_for_backward(idx)
        ctx.dims = (c, h, w, p, q, ks, stride, pad, x.shape)
        return out

    @staticmethod
    def backward(ctx, dout):
        from ..kernels import dispatch
        ext = dispatch.ext()
        (idx,) = ctx.saved_tensors
        c, h, w, p, q, ks, stride, pad, xshape = ctx.dims
        dout = dout.contiguous(memory_format=torch.channels_last)
        dx = torch.empty(xshape, device=dout.device, dtype=dout.dtype,
                         memory_format=torch.channels_last)
        ext.maxpool_cl_bwd(dout, idx, dx, c, h, w, p, q, ks, stride, pad)
        return dx, None, None, None


def max_pool2d(x, kernel_size, stride, padding=0):
    if _hip_ok(x) and _cl4(x):
        return MaxPool2dFn.apply(x, kernel_size, stride, padding)
    return F.max_pool2d(x, kernel_size=kernel_size, stride=stride,
                        padding=padding)


class GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        from ..kernels import dispatch
        ext = dispatch.ext()
        n, c, h, w = x.shape
        out = torch.empty(n, c, device=x.device, dtype=x.dtype)
        ext.gap_cl_fwd(x, out, c, h * w, n)
        ctx.dims = (c, h, w, x.shape)
        return out

    @staticmethod
    def backward(ctx, dout):
        from ..kernels import dispatch
        ext = dispatch.ext()
        c, h, w, xshape = ctx.dims
        dx = torch.empty(xshape, device=dout.device, dtype=dout.dtype,
                         memory_format=torch.channels_last)
        ext.gap_cl_bwd(dout.contiguous(), dx, c, h * w)
        return dx


def global_avg_pool(x):
    """(N, C, H, W) -> (N, C)."""
    if _hip_ok(x) and _cl4(x):
        return GlobalAvgPoolFn.apply(x)
    return F.adaptive_avg_pool2d(x, (1, 1)).reshape(x.shape[0], x.shape[1])


class MaxPool2dDWT(nn.Module):
    def __init__(self, kernel_size, stride, padding=0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x):
        return max_pool2d(x, self.kernel_size, self.stride, self.padding)
