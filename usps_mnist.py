#!/usr/bin/env python3
"""Digits (USPS<->MNIST) DWT domain-adaptation entrypoint.

CLI-compatible with the reference `usps_mnist.py:329-408`, plus:
  --loss {entropy,mec}   MEC needs the augmented target stream (BASELINE
                         configs 2-3 ask for DWT+MEC on digits)
  --synthetic            offline synthetic digits (no datasets in this env)
  --dtype, --whiten_mode, --data_root, checkpointing, DP-over-RCCL via
  torchrun environment variables.
"""
from __future__ import annotations

import argparse
import os

import torch
from torch.optim import lr_scheduler
from torch.utils.data import DataLoader, distributed as dist_data

from dwt_amd.data import (MNIST, USPS, Compose, Lambda, Normalize,
                          SyntheticDigits, ToTensor)
from dwt_amd.data.augment import random_affine_augmentation
from dwt_amd.engine.digits import test, train_digits_epoch
from dwt_amd.engine.meters import JsonlLogger
from dwt_amd.models import LeNet, checkpoint as ckpt
from dwt_amd.parallel import BucketedDataParallel, init_distributed_from_env
from dwt_amd.utils import seed_everything


def build_args(argv=None):
    parser = argparse.ArgumentParser(description='PyTorch DIAL example (dwt_amd)')
    parser.add_argument('--num_workers', default=2, type=int)
    parser.add_argument('--source_batch_size', type=int, default=32)
    parser.add_argument('--target_batch_size', type=int, default=32)
    parser.add_argument('--test_batch_size', type=int, default=100)
    parser.add_argument('--source', type=str, default='usps')
    parser.add_argument('--target', type=str, default='mnist')
    parser.add_argument('--epochs', type=int, default=120)
    parser.add_argument('--lr', type=float, default=0.001)
    parser.add_argument('--sgd_momentum', type=float, default=0.5)
    parser.add_argument('--running_momentum', type=float, default=0.1)
    parser.add_argument('--lambda_entropy_loss', type=float, default=0.1)
    parser.add_argument('--log_interval', type=int, default=100)
    parser.add_argument('--seed', type=int, default=1)
    parser.add_argument('--from_script', action='store_true')
    parser.add_argument('--run', default=0, type=int)
    parser.add_argument('--method', default='bn')
    parser.add_argument('--group_size', type=int, default=32)
    # --- dwt_amd extensions ---
    parser.add_argument('--loss', choices=['entropy', 'mec'], default='entropy')
    parser.add_argument('--synthetic', action='store_true',
                        help='synthetic digits (offline environments)')
    parser.add_argument('--synthetic_size', type=int, default=2048)
    parser.add_argument('--data_root', type=str, default='../data')
    parser.add_argument('--dtype', choices=['float32', 'bfloat16'], default='float32')
    parser.add_argument('--whiten_mode', choices=['chol', 'zca'], default='zca',
                        help='ZCA Newton-Schulz (primary) or Cholesky (reference-parity mode)')
    parser.add_argument('--checkpoint_path', type=str, default='')
    parser.add_argument('--resume', action='store_true')
    parser.add_argument('--metrics_jsonl', type=str, default='')
    parser.add_argument('--stats_mode', choices=['local', 'sync'], default='local',
                        help="per-rank norm statistics (stock-DDP-like) or "
                             "cross-rank synced batch stats")
    return parser.parse_args(argv)


def _digit_transform(name):
    if name == 'mnist':
        return Compose([ToTensor(), Normalize(mean=[0.1307], std=[0.3081])])
    return Compose([ToTensor(), Normalize(mean=[0.5], std=[0.5])])


def _aug_transform(base):
    return Compose([base, Lambda(random_affine_augmentation)])


def build_datasets(args):
    want_aug = args.loss == 'mec'
    if args.synthetic:
        t = _digit_transform('usps')
        aug = _aug_transform(t) if want_aug else None
        src_train = SyntheticDigits(args.synthetic_size, train=True, transform=t, seed=10)
        tgt_train = SyntheticDigits(args.synthetic_size, train=True, transform=t,
                                    transform_aug=aug, seed=20, shift=0.1)
        tgt_test = SyntheticDigits(max(args.synthetic_size // 4, 64), train=False,
                                   transform=t, seed=20, shift=0.1)
        return src_train, tgt_train, tgt_test

    def make(name, train, with_aug):
        t = _digit_transform(name)
        aug = _aug_transform(t) if (with_aug and train) else None
        if name == 'mnist':
            return MNIST(os.path.join(args.data_root, 'mnist'), train=train,
                         transform=t, transform_aug=aug)
        return USPS(os.path.join(args.data_root, 'usps'), train=train,
                    transform=t, transform_aug=aug)

    src_train = make(args.source, True, False)
    tgt_train = make(args.target, True, want_aug)
    tgt_test = make(args.target, False, False)
    return src_train, tgt_train, tgt_test


def main(argv=None):
    args = build_args(argv)
    if getattr(args, 'stats_mode', 'local') == 'sync':
        os.environ['DWT_AMD_STATS_SYNC'] = '1'
    assert args.source != args.target, "source and target datasets can not be the same"
    rank, world, local_rank = init_distributed_from_env()
    seed_everything(args.seed, rank)
    device = torch.device(f'cuda:{local_rank}' if torch.cuda.is_available() else 'cpu')
    dtype = torch.bfloat16 if args.dtype == 'bfloat16' else torch.float32

    src_train, tgt_train, tgt_test = build_datasets(args)
    samplers = {}
    if world > 1:
        samplers['src'] = dist_data.DistributedSampler(src_train, world, rank)
        samplers['tgt'] = dist_data.DistributedSampler(tgt_train, world, rank)
    src_loader = DataLoader(src_train, batch_size=args.source_batch_size,
                            shuffle='src' not in samplers, sampler=samplers.get('src'),
                            num_workers=args.num_workers, drop_last=True,
                            pin_memory=torch.cuda.is_available(),
                            persistent_workers=args.num_workers > 0)
    tgt_loader = DataLoader(tgt_train, batch_size=args.source_batch_size,
                            shuffle='tgt' not in samplers, sampler=samplers.get('tgt'),
                            num_workers=args.num_workers, drop_last=True,
                            pin_memory=torch.cuda.is_available(),
                            persistent_workers=args.num_workers > 0)
    test_loader = DataLoader(tgt_test, batch_size=args.test_batch_size,
                             shuffle=True, num_workers=args.num_workers)

    model = LeNet(group_size=args.group_size,
                  streams=3 if args.loss == 'mec' else 2,
                  whiten_mode=args.whiten_mode).to(device).to(dtype)
    from dwt_amd.ops.optim import FusedAdam
    optimizer = FusedAdam(model.parameters(), lr=args.lr, weight_decay=5e-4)
    sched = lr_scheduler.MultiStepLR(optimizer, milestones=[50, 80], gamma=0.1)

    start_epoch = 0
    if args.resume and args.checkpoint_path and os.path.exists(args.checkpoint_path):
        start_epoch = ckpt.load_training_state(args.checkpoint_path, model,
                                               optimizer, sched)
        print(f"Resumed from {args.checkpoint_path} at epoch {start_epoch}")

    ddp = BucketedDataParallel(model)
    logger = JsonlLogger(args.metrics_jsonl or None, rank)

    for epoch in range(start_epoch, args.epochs):
        if world > 1:
            samplers['src'].set_epoch(epoch)
            samplers['tgt'].set_epoch(epoch)
        sched.step()
        train_digits_epoch(args, model, device, src_loader, tgt_loader,
                           optimizer, epoch, args.lambda_entropy_loss,
                           loss_kind=args.loss, logger=logger,
                           grad_sync=ddp.sync if ddp.enabled else None)
        if world > 1:
            ddp.broadcast_buffers()
        if rank == 0:
            test(args, model, device, test_loader, logger=logger)
        if rank == 0 and args.checkpoint_path:
            ckpt.save_training_state(args.checkpoint_path, model, optimizer,
                                     sched, iteration=epoch + 1)
    logger.close()


if __name__ == '__main__':
    main()
