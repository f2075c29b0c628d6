"""Digits (USPS<->MNIST) train/eval loops.

Semantics per the reference (usps_mnist.py:281-327): concatenate the source
and target half-batches, forward once, classification loss on the source
half, entropy (or MEC, with a 3rd augmented stream) on the target, one
optimizer step.  Console strings match the reference's.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from ..ops import functional as Fdwt
from .meters import JsonlLogger, ThroughputMeter


def train_digits_epoch(args, model, device, source_loader, target_loader,
                       optimizer, epoch, lambda_loss, loss_kind="entropy",
                       logger: JsonlLogger = None, grad_sync=None):
    model.train()
    tp = ThroughputMeter()
    for batch_idx, (source, target) in enumerate(zip(source_loader, target_loader)):
        source_data, source_y = source[0], source[1]
        if loss_kind == "mec":
            target_data, target_dup = target[0], target[1]
            data = torch.cat((source_data, target_data, target_dup), dim=0)
        else:
            target_data = target[0]
            data = torch.cat((source_data, target_data), dim=0)
        data = data.to(device, non_blocking=True)
        source_y = source_y.to(device, non_blocking=True)
        if next(model.parameters()).dtype != data.dtype:
            data = data.to(next(model.parameters()).dtype)

        optimizer.zero_grad(set_to_none=True)
        output = model(data)
        parts = 3 if loss_kind == "mec" else 2
        chunks = torch.split(output, output.shape[0] // parts, dim=0)
        source_out = chunks[0]

        cls_loss = Fdwt.ce_loss(source_out, source_y)
        if loss_kind == "mec":
            aux = lambda_loss * Fdwt.mec_loss(chunks[1], chunks[2])
        else:
            aux = lambda_loss * Fdwt.entropy_loss(chunks[1])
        loss = cls_loss + aux
        loss.backward()
        if grad_sync is not None:
            grad_sync()
        optimizer.step()

        if batch_idx % args.log_interval == 0:
            ips = tp.tick(data.shape[0])
            print('Train Epoch: {} [{}/{} ({:.0f}%)]\tClassification Loss: {:.6f} \tEntropy Loss: {:.6f}'.format(
                epoch, batch_idx * len(target_data), len(source_loader.dataset),
                100. * batch_idx / len(source_loader), cls_loss.item(), aux.item()))
            if logger is not None:
                logger.log(kind="train", epoch=epoch, step=batch_idx,
                           cls_loss=cls_loss.item(), aux_loss=aux.item(),
                           imgs_per_sec=ips)


def test(args, model, device, target_test_loader, logger: JsonlLogger = None):
    model.eval()
    test_cls_loss = 0.0
    correct = 0
    dtype = next(model.parameters()).dtype
    with torch.no_grad():
        for data, target in target_test_loader:
            data = data.to(device).to(dtype)
            target = target.to(device)
            output = model(data).float()
            test_cls_loss += F.nll_loss(F.log_softmax(output, dim=1), target,
                                        reduction="sum").item()
            pred = F.softmax(output, dim=1).max(1, keepdim=True)[1]
            correct += pred.eq(target.view_as(pred)).sum().item()

    n = len(target_test_loader.dataset)
    test_cls_loss /= n
    acc = 100.0 * correct / n
    print('\nTest set: Classification loss: {:.4f}, Accuracy: {}/{} ({:.2f}%)\n'.format(
        test_cls_loss, correct, n, acc))
    if logger is not None:
        logger.log(kind="test", loss=test_cls_loss, acc=acc)
    return acc
