#!/usr/bin/env python3
"""Digits (LeNet-DWT) training-step throughput — BASELINE.json configs 2-3
(USPS<->MNIST DWT+MEC bf16 on 1xMI355X; the direction only changes data,
not compute, so one measurement covers both).

Prints one JSON line per mode (entropy 2-stream / MEC 3-stream).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1024, help="per-domain batch")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--group_size", type=int, default=4)
    ap.add_argument("--dtype", choices=["bfloat16", "float32"],
                    default="bfloat16" if torch.cuda.is_available() else "float32")
    args = ap.parse_args()
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if args.dtype == "bfloat16" else torch.float32
    if device.type == "cpu":
        args.batch = min(args.batch, 16)

    from dwt_amd.models import LeNet
    from dwt_amd.ops import functional as Fdwt
    from dwt_amd.ops.optim import FusedAdam

    for loss_kind, streams in (("entropy", 2), ("mec", 3)):
        torch.manual_seed(0)
        model = LeNet(group_size=args.group_size, streams=streams)
        model = model.to(device).to(dtype).train()
        opt = FusedAdam(model.parameters(), lr=1e-3, weight_decay=5e-4)
        b = args.batch
        data = torch.randn(streams * b, 1, 28, 28, device=device, dtype=dtype)
        labels = torch.randint(0, 10, (b,), device=device)

        def step():
            opt.zero_grad(set_to_none=True)
            out = model(data)
            chunks = torch.split(out, b, dim=0)
            loss = F.nll_loss(F.log_softmax(chunks[0].float(), 1), labels)
            if loss_kind == "mec":
                loss = loss + 0.1 * Fdwt.mec_loss(chunks[1], chunks[2])
            else:
                loss = loss + 0.1 * Fdwt.entropy_loss(chunks[1])
            loss.backward()
            opt.step()

        for _ in range(args.warmup):
            step()
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            step()
        if device.type == "cuda":
            torch.cuda.synchronize()
        el = time.perf_counter() - t0
        print(json.dumps({
            "metric": "digits_imgs_per_sec",
            "value": streams * b * args.steps / el,
            "unit": "images/sec",
            "ms_per_step": 1000 * el / args.steps,
            "config": {"model": f"lenet-dwt-{loss_kind}", "streams": streams,
                       "per_domain_batch": b, "group_size": args.group_size,
                       "dtype": args.dtype},
        }))


if __name__ == "__main__":
    main()
