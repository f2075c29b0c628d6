#!/usr/bin/env python3
"""Office-Home DWT-MEC (ResNet50) entrypoint.

CLI-compatible with the reference `resnet50_dwt_mec_officehome.py:495-603`,
plus offline-synthetic mode, bf16, checkpoint save/resume, and data
parallelism over RCCL/xGMI via torchrun env vars.
"""
from __future__ import annotations

import argparse
import os

import torch
from torch.utils.data import DataLoader, distributed as dist_data

from dwt_amd.data import (Compose, ImageFolder, Lambda, Normalize, RandomCrop,
                          RandomHorizontalFlip, Resize, SyntheticOfficeHome,
                          ToTensor)
from dwt_amd.data.augment import gaussian_blur, random_affine_augmentation
from dwt_amd.engine.meters import JsonlLogger
from dwt_amd.engine.officehome import train_infinite_collect_stats
from dwt_amd.models import resnet50, Bottleneck, ResNetDWT
from dwt_amd.parallel import BucketedDataParallel, init_distributed_from_env
from dwt_amd.utils import seed_everything


def build_args(argv=None):
    parser = argparse.ArgumentParser(description='PyTorch DWT-MEC OfficeHome (dwt_amd)')
    parser.add_argument('--num_workers', default=2, type=int)
    parser.add_argument('--source_batch_size', type=int, default=18)
    parser.add_argument('--target_batch_size', type=int, default=18)
    parser.add_argument('--test_batch_size', type=int, default=10)
    parser.add_argument('--s_dset_path', type=str,
                        default='../data/OfficeHomeDataset_10072016/Art')
    parser.add_argument('--t_dset_path', type=str,
                        default='../data/OfficeHomeDataset_10072016/Clipart')
    parser.add_argument('--resnet_path', type=str,
                        default='../data/models/model_best_gr_4.pth.tar')
    parser.add_argument('--img_resize', type=int, default=256)
    parser.add_argument('--img_crop_size', type=int, default=224)
    parser.add_argument('--num_iters', type=int, default=10000)
    parser.add_argument('--check_acc_step', type=int, default=100)
    parser.add_argument('--lr_change_step', type=int, default=1000)
    parser.add_argument('--lr', type=float, default=1e-2)
    parser.add_argument('--num_classes', type=int, default=65)
    parser.add_argument('--sgd_momentum', type=float, default=0.5)
    parser.add_argument('--running_momentum', type=float, default=0.1)
    parser.add_argument('--lambda_mec_loss', type=float, default=0.1)
    parser.add_argument('--log_interval', type=int, default=10)
    parser.add_argument('--seed', type=int, default=1)
    # --- dwt_amd extensions ---
    parser.add_argument('--synthetic', action='store_true',
                        help='synthetic Office-Home-shaped data + random init')
    parser.add_argument('--synthetic_size', type=int, default=2048)
    parser.add_argument('--group_size', type=int, default=4)
    parser.add_argument('--dtype', choices=['float32', 'bfloat16'],
                        default='bfloat16' if torch.cuda.is_available() else 'float32')
    parser.add_argument('--whiten_mode', choices=['chol', 'zca'], default='zca',
                        help='ZCA Newton-Schulz (primary) or Cholesky (reference-parity mode)')
    parser.add_argument('--checkpoint_path', type=str, default='')
    parser.add_argument('--checkpoint_every', type=int, default=1000)
    parser.add_argument('--resume', action='store_true')
    parser.add_argument('--metrics_jsonl', type=str, default='')
    parser.add_argument('--stats_mode', choices=['local', 'sync'], default='local',
                        help="per-rank norm statistics (stock-DDP-like) or "
                             "cross-rank synced batch stats")
    parser.add_argument('--stats_passes', type=int, default=10,
                        help='target-stats re-estimation passes before final test')
    parser.add_argument('--gpu_augment', action='store_true',
                        help='build the duplicate MEC target view on-device '
                             '(batched flip+affine) instead of in the CPU '
                             'loader workers — keeps 8-GPU runs unstarved')
    return parser.parse_args(argv)


def build_loaders(args, rank, world):
    aug_in_loader = not getattr(args, 'gpu_augment', False)
    if args.synthetic:
        src = SyntheticOfficeHome(args.synthetic_size, args.num_classes,
                                  args.img_crop_size, seed=1)
        tgt = SyntheticOfficeHome(args.synthetic_size, args.num_classes,
                                  args.img_crop_size,
                                  transform_aug=aug_in_loader, seed=2)
        tgt_test = SyntheticOfficeHome(max(args.synthetic_size // 4, 64),
                                       args.num_classes, args.img_crop_size, seed=2)
    else:
        data_transform = Compose([
            Resize((args.img_resize, args.img_resize)),
            RandomCrop((args.img_crop_size, args.img_crop_size)),
            ToTensor(),
            Normalize(mean=[0.485, 0.456, 0.406], std=[0.229, 0.224, 0.225]),
        ])
        data_transform_dup = Compose([
            Resize((args.img_resize, args.img_resize)),
            RandomCrop((args.img_crop_size, args.img_crop_size)),
            RandomHorizontalFlip(),
            ToTensor(),
            Lambda(random_affine_augmentation),
            Lambda(gaussian_blur),
            Normalize(mean=[0.485, 0.456, 0.406], std=[0.229, 0.224, 0.225]),
        ])
        src = ImageFolder(root=args.s_dset_path, transform=data_transform)
        tgt = ImageFolder(root=args.t_dset_path, transform=data_transform,
                          transform_aug=data_transform_dup if aug_in_loader
                          else None)
        tgt_test = ImageFolder(root=args.t_dset_path, transform=data_transform)

    samplers = {}
    if world > 1:
        samplers['src'] = dist_data.DistributedSampler(src, world, rank)
        samplers['tgt'] = dist_data.DistributedSampler(tgt, world, rank)
    src_loader = DataLoader(src, batch_size=args.source_batch_size,
                            shuffle='src' not in samplers, sampler=samplers.get('src'),
                            num_workers=args.num_workers, drop_last=True,
                            pin_memory=torch.cuda.is_available(),
                            persistent_workers=args.num_workers > 0)
    # note: the reference uses source_batch_size for the target loader too
    #  — kept for parity
    tgt_loader = DataLoader(tgt, batch_size=args.source_batch_size,
                            shuffle='tgt' not in samplers, sampler=samplers.get('tgt'),
                            num_workers=args.num_workers, drop_last=True,
                            pin_memory=torch.cuda.is_available(),
                            persistent_workers=args.num_workers > 0)
    test_loader = DataLoader(tgt_test, batch_size=args.test_batch_size,
                             shuffle=True, num_workers=args.num_workers)
    return src_loader, tgt_loader, test_loader


def main(argv=None):
    args = build_args(argv)
    if getattr(args, 'stats_mode', 'local') == 'sync':
        os.environ['DWT_AMD_STATS_SYNC'] = '1'
    rank, world, local_rank = init_distributed_from_env()
    seed_everything(args.seed, rank)
    device = torch.device(f'cuda:{local_rank}' if torch.cuda.is_available() else 'cpu')
    dtype = torch.bfloat16 if args.dtype == 'bfloat16' else torch.float32

    if args.synthetic or not os.path.exists(args.resnet_path):
        if not args.synthetic:
            print(f"resnet_path {args.resnet_path} not found -> random init")
        model = ResNetDWT(Bottleneck, [3, 4, 6, 3], None,
                          num_classes=args.num_classes, group_size=args.group_size,
                          whiten_mode=args.whiten_mode)
    else:
        model = resnet50(args.resnet_path, device, num_classes=args.num_classes,
                         group_size=args.group_size, whiten_mode=args.whiten_mode)
    model = model.to(device).to(dtype)

    final_layer_params, rest = [], []
    for name, param in model.named_parameters():
        (final_layer_params if name.startswith('fc_out') else rest).append(param)
    from dwt_amd.ops.optim import FusedSGD
    optimizer = FusedSGD([
        {'params': rest},
        {'params': final_layer_params, 'lr': args.lr},
    ], lr=args.lr * 0.1, momentum=0.9, weight_decay=5e-4)

    start_iter = 0
    if args.resume and args.checkpoint_path and os.path.exists(args.checkpoint_path):
        from dwt_amd.models import checkpoint as ckpt
        start_iter = ckpt.load_training_state(args.checkpoint_path, model, optimizer)
        print(f"Resumed from {args.checkpoint_path} at iter {start_iter}")

    ddp = BucketedDataParallel(model)
    logger = JsonlLogger(args.metrics_jsonl or None, rank)
    src_loader, tgt_loader, test_loader = build_loaders(args, rank, world)

    train_infinite_collect_stats(
        args=args, model=model, device=device,
        source_train_loader=src_loader, target_train_loader=tgt_loader,
        optimizer=optimizer, lambda_mec_loss=args.lambda_mec_loss,
        target_test_loader=test_loader, logger=logger,
        grad_sync=ddp.sync if ddp.enabled else None, start_iter=start_iter,
        checkpoint_path=args.checkpoint_path or None,
        checkpoint_every=args.checkpoint_every, stats_passes=args.stats_passes)
    logger.close()


if __name__ == '__main__':
    main()
