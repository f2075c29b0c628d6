#!/usr/bin/env python3
"""Whitening/BN site microbenchmark: achieved HBM bandwidth per pass vs the
~6.3 TB/s MI355X ceiling, at R50 site shapes (per-domain batch B).

Traffic model per site (bf16 elements, bytes = 2*numel unless noted):
  whiten fwd  : stats reads x; apply reads x, writes out      -> 3x
  whiten bwd  : reduce reads x,dy[,out]; apply reads x,dy[,out], writes dx
  bn likewise.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=512, help="per-domain batch")
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    from dwt_amd.kernels.hip_ops import _HipWhitenMulti, _HipBatchNormMulti

    b = args.batch
    # (kind, C, HW-side) — R50 norm sites (per bottleneck appearance count folded out)
    sites = [
        ("wh", 64, 112), ("wh", 64, 56), ("wh", 256, 56),
        ("bn", 128, 56), ("bn", 128, 28), ("bn", 512, 28),
        ("bn", 256, 14), ("bn", 1024, 14), ("bn", 512, 7), ("bn", 2048, 7),
    ]
    lines = ["| site | fwd ms | fwd TB/s | bwd ms | bwd TB/s |", "|---|---|---|---|---|"]
    for kind, c, s in sites:
        x = torch.randn(3 * b, c, s, s, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last).requires_grad_(True)
        gamma = torch.randn(c, 1, 1, device=dev, dtype=torch.bfloat16, requires_grad=True)
        beta = torch.randn(c, 1, 1, device=dev, dtype=torch.bfloat16, requires_grad=True)
        nbytes = x.numel() * 2

        if kind == "wh":
            cfg = dict(parts=3, num_groups=c // 4, eps=1e-3, momentum=0.1,
                       training=True, mode="chol", relu=True)
            fwd = lambda: _HipWhitenMulti.apply(x, gamma, beta, None, None, cfg)
        else:
            cfg = dict(parts=3, eps=1e-5, momentum=0.1, training=True, relu=True)
            fwd = lambda: _HipBatchNormMulti.apply(x, gamma, beta, None, None, cfg)

        t_f = timeit(fwd)
        out = fwd()
        g = torch.randn_like(out)
        def bwd():
            x.grad = None
            out.backward(g, retain_graph=True)
        t_b = timeit(bwd, iters=10, warmup=3)

        fwd_traffic = 3 * nbytes            # stats read + apply read + write
        # reduce reads x,dy; apply reads x,dy + writes dx (the ReLU mask is
        # recomputed — the stored out is no longer re-read in backward)
        bwd_traffic = 5 * nbytes
        lines.append(f"| {kind} C={c} {s}x{s} B={3*b} | {t_f*1e3:.3f} | "
                     f"{fwd_traffic/t_f/1e12:.2f} | {t_b*1e3:.3f} | "
                     f"{bwd_traffic/t_b/1e12:.2f} |")
        print(lines[-1], flush=True)

    if args.out:
        with open(args.out, "w") as f:
            f.write("# Norm-site achieved bandwidth (vs ~6.3 TB/s ceiling)\n\n"
                    + "\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
