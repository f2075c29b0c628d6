#!/usr/bin/env python3
"""Per-shape microbenchmark: hand-written MFMA implicit-GEMM conv/GEMM vs
MIOpen/rocBLAS on the R50 forward shapes. Writes a markdown table.

Run on the GPU box:  python benchmarks/conv_bench.py [--out profiles/conv_bench.md]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

R50_SHAPES = [
    # (name, N, Cin, H, W, Cout, K, stride, pad)
    ("stem7x7", 192, 3, 224, 224, 64, 7, 2, 3),
    ("l1.conv1", 192, 64, 56, 56, 64, 1, 1, 0),
    ("l1.conv2", 192, 64, 56, 56, 64, 3, 1, 1),
    ("l1.conv3", 192, 64, 56, 56, 256, 1, 1, 0),
    ("l2.conv1", 192, 256, 56, 56, 128, 1, 1, 0),
    ("l2.conv2s", 192, 128, 56, 56, 128, 3, 2, 1),
    ("l2.conv3", 192, 128, 28, 28, 512, 1, 1, 0),
    ("l3.conv2", 192, 256, 28, 28, 256, 3, 2, 1),
    ("l3.conv3", 192, 256, 14, 14, 1024, 1, 1, 0),
    ("l4.conv2", 192, 512, 14, 14, 512, 3, 2, 1),
    ("l4.conv3", 192, 512, 7, 7, 2048, 1, 1, 0),
]

GEMM_SHAPES = [
    ("fc_out", 192, 2048, 65),
    ("sq2048", 2048, 2048, 2048),
    ("sq4096", 4096, 4096, 4096),
]


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
    ap.add_argument("--out", default="profiles/conv_bench.md")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--batch", type=int, default=192,
                    help="N (the flagship bench step runs N=1536 = 3x512)")
    ap.add_argument("--only", default="",
                    help="comma-separated substring filter on shape names")
    ap.add_argument("--passes", default="fwd,dgrad,wgrad")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    from dwt_amd.ops.mfma import conv2d_fwd, mfma_gemm
    dev = torch.device("cuda:0")
    lines = ["# MFMA implicit-GEMM conv/GEMM vs MIOpen/rocBLAS (bf16, NHWC)",
             "",
             "| shape | M x N x K | GFLOP | ours ms | lib ms | ours TF | lib TF | ratio |",
             "|---|---|---|---|---|---|---|---|"]

    from dwt_amd.ops.mfma import conv2d_dgrad, conv2d_wgrad

    def lib_dgrad(g, x, wt, stride, pad):
        return torch.ops.aten.convolution_backward(
            g, x, wt, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [True, False, False])[0]

    def lib_wgrad(g, x, wt, stride, pad):
        return torch.ops.aten.convolution_backward(
            g, x, wt, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [False, True, False])[1]

    only = [t for t in args.only.split(",") if t]
    passes_on = args.passes.split(",")
    for name, n, cin, h, w, cout, k, stride, pad in R50_SHAPES:
        n = args.batch
        if only and not any(t in name for t in only):
            continue
        x = torch.randn(n, cin, h, w, device=dev).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        wt = (torch.randn(cout, cin, k, k, device=dev) * 0.05).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        p = (h + 2 * pad - k) // stride + 1
        m = n * p * p
        kk = k * k * cin
        g = torch.randn(n, cout, p, p, device=dev).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        gflop = 2.0 * m * cout * kk / 1e9
        passes = [
            ("fwd", lambda: conv2d_fwd(x, wt, stride=stride, padding=pad),
             lambda: F.conv2d(x, wt, stride=stride, padding=pad)),
            ("dgrad", lambda: conv2d_dgrad(g, wt, x.shape, stride=stride, padding=pad),
             lambda: lib_dgrad(g, x, wt, stride, pad)),
            ("wgrad", lambda: conv2d_wgrad(g, x, tuple(wt.shape), stride=stride, padding=pad),
             lambda: lib_wgrad(g, x, wt, stride, pad)),
        ]
        for pname, ours_fn, lib_fn in passes:
            if pname not in passes_on:
                continue
            if pname == "dgrad" and name == "stem7x7":
                continue  # stem never needs dx
            t_ours = timeit(ours_fn, args.iters)
            t_lib = timeit(lib_fn, args.iters)
            lines.append(f"| {name}:{pname} | {m}x{cout}x{kk} | {gflop:.1f} | "
                         f"{t_ours*1e3:.3f} | {t_lib*1e3:.3f} | "
                         f"{gflop/t_ours/1e3:.0f} | {gflop/t_lib/1e3:.0f} | "
                         f"{t_lib/t_ours:.2f}x |")
            print(lines[-1], flush=True)

    for name, m, k, n in GEMM_SHAPES:
        a = torch.randn(m, k, device=dev).to(torch.bfloat16)
        bt = torch.randn(n, k, device=dev).to(torch.bfloat16)
        b = bt.t().contiguous()
        gflop = 2.0 * m * n * k / 1e9
        t_ours = timeit(lambda: mfma_gemm(a, bt), args.iters)
        t_lib = timeit(lambda: a @ b, args.iters)
        lines.append(f"| gemm:{name} | {m}x{n}x{k} | {gflop:.1f} | "
                     f"{t_ours*1e3:.3f} | {t_lib*1e3:.3f} | "
                     f"{gflop/t_ours/1e3:.0f} | {gflop/t_lib/1e3:.0f} | "
                     f"{t_lib/t_ours:.2f}x |")
        print(lines[-1], flush=True)

    with open(args.out, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
