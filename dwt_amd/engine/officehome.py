"""Office-Home (DWT-ResNet50 + MEC) train/eval loops.

Semantics per the reference (resnet50_dwt_mec_officehome.py:380-464):
infinite-iterator loop over (source, target, target_aug) thirds, CE on the
source third + lambda * MEC on the two target views, SGD step, periodic eval;
after training, a target-stats re-estimation pass (`eval_pass_collect_stats`)
then the final test.  Adds (new vs reference): checkpoint save/resume,
JSONL metrics, data-parallel gradient sync hook.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch.optim import lr_scheduler

from ..models import checkpoint as ckpt
from ..ops import functional as Fdwt
from .meters import JsonlLogger, ThroughputMeter


def eval_pass_collect_stats(args, model, device, target_test_loader, passes=10):
    """Re-estimate EMA stats with target data in all three streams
    (resnet50_dwt_mec_officehome.py:380-389)."""
    model.train(mode=True)
    dtype = next(model.parameters()).dtype
    with torch.no_grad():
        for i in range(passes):
            print("Pass {} ...".format(i))
            for data, _ in target_test_loader:
                data = torch.cat((data, data, data), dim=0).to(device).to(dtype)
                model(data)


def train_infinite_collect_stats(args, model, device, source_train_loader,
                                 target_train_loader, optimizer, lambda_mec_loss,
                                 target_test_loader, logger: JsonlLogger = None,
                                 grad_sync=None, start_iter=0,
                                 checkpoint_path=None, checkpoint_every=0,
                                 stats_passes=10):
    source_iter = iter(source_train_loader)
    target_iter = iter(target_train_loader)
    exp_lr_scheduler = lr_scheduler.MultiStepLR(optimizer, milestones=[6000], gamma=0.1)
    for _ in range(start_iter):  # fast-forward on resume
        exp_lr_scheduler.step()
    dtype = next(model.parameters()).dtype
    tp = ThroughputMeter()

    progress = {"iter": start_iter}
    try:
        return _train_loop(args, model, device, source_iter, target_iter,
                           source_train_loader, target_train_loader,
                           optimizer, exp_lr_scheduler, lambda_mec_loss,
                           target_test_loader, logger, grad_sync, start_iter,
                           checkpoint_path, checkpoint_every, stats_passes,
                           dtype, tp, progress)
    except Exception:
        # failure handling: persist an emergency checkpoint so a
        # torchrun restart can --resume instead of losing the run
        if checkpoint_path:
            try:
                ckpt.save_training_state(checkpoint_path + ".emergency", model,
                                         optimizer, exp_lr_scheduler,
                                         iteration=progress["iter"])
                print(f"emergency checkpoint written: {checkpoint_path}.emergency")
            except Exception as exc:  # keep the original error primary
                print(f"emergency checkpoint failed: {exc}")
        raise


def _train_loop(args, model, device, source_iter, target_iter,
                source_train_loader, target_train_loader, optimizer,
                exp_lr_scheduler, lambda_mec_loss, target_test_loader, logger,
                grad_sync, start_iter, checkpoint_path, checkpoint_every,
                stats_passes, dtype, tp, progress=None):
    for i in range(start_iter, args.num_iters):
        if progress is not None:
            progress["iter"] = i
        model.train()
        # scheduler stepped before the optimizer on purpose: reference
        # behavior (resnet50_dwt_mec_officehome.py:403 — SURVEY quirk #9)
        exp_lr_scheduler.step()
        try:
            source_data, source_y = next(source_iter)
        except StopIteration:
            source_iter = iter(source_train_loader)
            source_data, source_y = next(source_iter)
        try:
            tb = next(target_iter)
        except StopIteration:
            target_iter = iter(target_train_loader)
            tb = next(target_iter)

        gpu_aug = getattr(args, "gpu_augment", False)
        if gpu_aug:
            # loader yields (data, label); the duplicate MEC view is built
            # on-device (batched flip + affine — data/augment.py)
            from ..data import augment
            target_data = tb[0]
            pair = torch.cat((source_data, target_data), dim=0)
            pair = pair.to(device, non_blocking=True).to(dtype)
            tdev = pair[source_data.shape[0]:]
            data = torch.cat((pair, augment.gpu_target_views(tdev)), dim=0)
        else:
            target_data, target_data_dup, _ = tb
            data = torch.cat((source_data, target_data, target_data_dup), dim=0)
            data = data.to(device, non_blocking=True).to(dtype)
        source_y = source_y.to(device, non_blocking=True)

        optimizer.zero_grad(set_to_none=True)
        output = model(data)
        source_out, target_out, target_out_dup = torch.split(
            output, output.shape[0] // 3, dim=0)

        cls_loss = Fdwt.ce_loss(source_out, source_y)
        mec_loss = lambda_mec_loss * Fdwt.mec_loss(target_out, target_out_dup)
        loss = cls_loss + mec_loss
        loss.backward()
        if grad_sync is not None:
            grad_sync()
        optimizer.step()

        if i % args.log_interval == 0:
            ips = tp.tick(data.shape[0])
            print('Train Iter: [{}/{}]\tClassification Loss: {:.6f} \t MEC Loss: {:.6f}'.format(
                i, args.num_iters, cls_loss.item(), mec_loss.item()))
            if logger is not None:
                logger.log(kind="train", step=i, cls_loss=cls_loss.item(),
                           mec_loss=mec_loss.item(), imgs_per_sec=ips)

        if (i + 1) % args.check_acc_step == 0:
            test(args, model, device, target_test_loader, logger=logger)

        if checkpoint_path and checkpoint_every and (i + 1) % checkpoint_every == 0:
            ckpt.save_training_state(checkpoint_path, model, optimizer,
                                     exp_lr_scheduler, iteration=i + 1)

    print("Training is complete...")
    print("Running a bunch of forward passes to estimate the population statistics of target...")
    eval_pass_collect_stats(args, model, device, target_test_loader,
                            passes=stats_passes)
    print("Finally computing the precision on the test set...")
    return test(args, model, device, target_test_loader, logger=logger)


def test(args, model, device, target_test_loader, logger: JsonlLogger = None):
    model.eval()
    test_loss = 0.0
    correct = 0
    dtype = next(model.parameters()).dtype
    with torch.no_grad():
        for data, target in target_test_loader:
            data = data.to(device).to(dtype)
            target = target.to(device)
            output = model(data).float()
            test_loss += F.nll_loss(F.log_softmax(output, dim=1), target,
                                    reduction="sum").item()
            pred = F.softmax(output, dim=1).max(1, keepdim=True)[1]
            correct += pred.eq(target.view_as(pred)).sum().item()

    n = len(target_test_loader.dataset)
    test_loss /= n
    acc = 100.0 * correct / n
    print('\nTest set: Average loss: {:.4f}, Accuracy: {}/{} ({:.2f}%)\n'.format(
        test_loss, correct, n, acc))
    if logger is not None:
        logger.log(kind="test", loss=test_loss, acc=acc)
    return acc
