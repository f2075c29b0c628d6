#!/usr/bin/env python3
"""Office-Home 12-pair domain-adaptation sweep (BASELINE.json config 5).

Runs the DWT-MEC ResNet50 trainer over every ordered pair of the four
Office-Home domains (Art, Clipart, Product, Real World) and reports the
per-pair and average target top-1.  With real data:

    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        benchmarks/officehome_sweep.py --data_root /data/OfficeHomeDataset_10072016 \
        --resnet_path /data/models/model_best_gr_4.pth.tar

Offline (no dataset in this environment) use --synthetic, which exercises the
full sweep plumbing on synthetic per-domain datasets.
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

DOMAINS = ["Art", "Clipart", "Product", "Real World"]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--data_root", default="../data/OfficeHomeDataset_10072016")
    ap.add_argument("--resnet_path", default="../data/models/model_best_gr_4.pth.tar")
    ap.add_argument("--num_iters", type=int, default=10000)
    ap.add_argument("--synthetic", action="store_true")
    ap.add_argument("--synthetic_size", type=int, default=256)
    ap.add_argument("--out", default="officehome_sweep.json")
    ap.add_argument("--pairs", default="",
                    help="comma list like 'Art:Clipart,Product:Art' (default all 12)")
    args, extra = ap.parse_known_args()
    args.extra = extra
    wanted = None
    if args.pairs:
        wanted = {tuple(p.split(":")) for p in args.pairs.split(",")}

    import resnet50_dwt_mec_officehome as oh
    from dwt_amd.engine import officehome as engine

    results = {}
    for src in DOMAINS:
        for tgt in DOMAINS:
            if src == tgt or (wanted is not None and (src, tgt) not in wanted):
                continue
            pair = f"{src}->{tgt}"
            print(f"===== {pair} =====", flush=True)
            argv = ["--num_iters", str(args.num_iters),
                    "--s_dset_path", os.path.join(args.data_root, src),
                    "--t_dset_path", os.path.join(args.data_root, tgt),
                    "--resnet_path", args.resnet_path] + args.extra
            if args.synthetic:
                argv += ["--synthetic", "--synthetic_size", str(args.synthetic_size)]
            # capture the final accuracy by running the pair's training
            acc_holder = {}
            orig = oh.train_infinite_collect_stats

            def wrapper(*a, **kw):
                acc_holder["acc"] = orig(*a, **kw)
                return acc_holder["acc"]

            oh.train_infinite_collect_stats = wrapper
            try:
                oh.main(argv)
            finally:
                oh.train_infinite_collect_stats = orig
            results[pair] = acc_holder.get("acc")
            print(f"{pair}: {results[pair]}", flush=True)

    accs = [v for v in results.values() if v is not None]
    results["average"] = sum(accs) / max(len(accs), 1)
    rank = int(os.environ.get("RANK", "0"))
    if rank == 0:
        with open(args.out, "w") as f:
            json.dump(results, f, indent=2)
        print(json.dumps(results, indent=2))


if __name__ == "__main__":
    main()
