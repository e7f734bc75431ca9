"""Train/evaluate loops — reference bin/utils.py:183-275 semantics.

train(): zero_grad -> model(input, age) -> loss -> NaN tripwire (utils.py:206)
-> backward -> step, with AverageMeter timing/loss/accuracy and periodic
prints. evaluate(): no_grad loop collecting (y_true, y_pred) pairs.
Device-agnostic: the loaders yield (input, age, target) like the reference's
TensorDataset (utils.py:365-384).
"""

from __future__ import annotations

import time
from typing import List, Tuple

import numpy as np
import torch

from tskd_amd.train.metrics import AverageMeter, compute_batch_accuracy


def train(model, device, data_loader, criterion, optimizer, epoch,
          print_freq: int = 10) -> Tuple[float, float]:
    batch_time = AverageMeter()
    data_time = AverageMeter()
    losses = AverageMeter()
    accuracy = AverageMeter()

    model.train()
    end = time.time()
    for i, (inp, age, target) in enumerate(data_loader):
        data_time.update(time.time() - end)
        inp, age, target = inp.to(device), age.to(device), target.to(device)
        optimizer.zero_grad()
        output = model(inp, age)
        loss = criterion(output, target)
        assert not np.isnan(loss.item()), "model diverged with loss = NaN"
        loss.backward()
        optimizer.step()
        batch_time.update(time.time() - end)
        end = time.time()
        losses.update(loss.item(), target.size(0))
        accuracy.update(compute_batch_accuracy(output, target).item(),
                        target.size(0))
        if i % print_freq == 0:
            print(f"Epoch: [{epoch}][{i}/{len(data_loader)}]\t"
                  f"Time {batch_time.val:.3f} ({batch_time.avg:.3f})\t"
                  f"Data {data_time.val:.3f} ({data_time.avg:.3f})\t"
                  f"Loss {losses.val:.4f} ({losses.avg:.4f})\t"
                  f"Accuracy {accuracy.val:.3f} ({accuracy.avg:.3f})")
    return losses.avg, accuracy.avg


def evaluate(model, device, data_loader, criterion,
             print_freq: int = 10) -> Tuple[float, float, List[Tuple[int, int]]]:
    batch_time = AverageMeter()
    losses = AverageMeter()
    accuracy = AverageMeter()
    results: List[Tuple[int, int]] = []

    model.eval()
    with torch.no_grad():
        end = time.time()
        for i, (inp, age, target) in enumerate(data_loader):
            inp, age, target = (inp.to(device), age.to(device),
                                target.to(device))
            output = model(inp, age)
            loss = criterion(output, target)
            batch_time.update(time.time() - end)
            end = time.time()
            losses.update(loss.item(), target.size(0))
            accuracy.update(compute_batch_accuracy(output, target).item(),
                            target.size(0))
            y_true = target.detach().cpu().numpy().flatten().tolist()
            y_pred = torch.sigmoid(output).round().detach().cpu().numpy() \
                .flatten().tolist()
            results.extend(zip(y_true, y_pred))
            if i % print_freq == 0:
                print(f"Test: [{i}/{len(data_loader)}]\t"
                      f"Time {batch_time.val:.3f} ({batch_time.avg:.3f})\t"
                      f"Loss {losses.val:.4f} ({losses.avg:.4f})\t"
                      f"Accuracy {accuracy.val:.3f} ({accuracy.avg:.3f})")
    return losses.avg, accuracy.avg, results
