"""Train/evaluate loops — reference bin/utils.py:183-275 semantics.

train(): zero_grad -> model(input, age) -> loss -> NaN tripwire (utils.py:206)
-> backward -> step, with per-batch timing/loss/accuracy meters and periodic
progress lines. evaluate(): no_grad loop collecting (y_true, y_pred) pairs.
Device-agnostic: the loaders yield (input, age, target) like the reference's
TensorDataset (utils.py:365-384). The required parity is behavioral (meter
cadence, NaN assert, returned averages + result pairs); the progress-line
formatting is our own.
"""

from __future__ import annotations

import time
from typing import List, Tuple

import numpy as np
import torch

from tskd_amd.train.metrics import AverageMeter, compute_batch_accuracy


def _progress(tag: str, i: int, n: int, meters: dict) -> str:
    parts = [f"[{tag} {i + 1}/{n}]"]
    for name, m in meters.items():
        parts.append(f"{name}={m.val:.4g} avg={m.avg:.4g}")
    return "  ".join(parts)


def train(model, device, data_loader, criterion, optimizer, epoch,
          print_freq: int = 10) -> Tuple[float, float]:
    t_step = AverageMeter()
    t_data = AverageMeter()
    m_loss = AverageMeter()
    m_acc = AverageMeter()

    model.train()
    end = time.time()
    for i, (inp, age, target) in enumerate(data_loader):
        t_data.update(time.time() - end)
        inp, age, target = inp.to(device), age.to(device), target.to(device)
        optimizer.zero_grad()
        output = model(inp, age)
        loss = criterion(output, target)
        assert not np.isnan(loss.item()), "model diverged with loss = NaN"
        loss.backward()
        optimizer.step()
        t_step.update(time.time() - end)
        end = time.time()
        m_loss.update(loss.item(), target.size(0))
        m_acc.update(compute_batch_accuracy(output, target).item(),
                     target.size(0))
        if i % print_freq == 0:
            print(_progress(f"train e{epoch}", i, len(data_loader),
                            {"loss": m_loss, "acc": m_acc,
                             "step_s": t_step, "data_s": t_data}))
    return m_loss.avg, m_acc.avg


def evaluate(model, device, data_loader, criterion,
             print_freq: int = 10) -> Tuple[float, float, List[Tuple[int, int]]]:
    t_step = AverageMeter()
    m_loss = AverageMeter()
    m_acc = AverageMeter()
    results: List[Tuple[int, int]] = []

    model.eval()
    with torch.no_grad():
        end = time.time()
        for i, (inp, age, target) in enumerate(data_loader):
            inp, age, target = (inp.to(device), age.to(device),
                                target.to(device))
            output = model(inp, age)
            loss = criterion(output, target)
            t_step.update(time.time() - end)
            end = time.time()
            m_loss.update(loss.item(), target.size(0))
            m_acc.update(compute_batch_accuracy(output, target).item(),
                         target.size(0))
            y_true = target.detach().cpu().numpy().flatten().tolist()
            y_pred = torch.sigmoid(output).round().detach().cpu().numpy() \
                .flatten().tolist()
            results.extend(zip(y_true, y_pred))
            if i % print_freq == 0:
                print(_progress("eval", i, len(data_loader),
                                {"loss": m_loss, "acc": m_acc,
                                 "step_s": t_step}))
    return m_loss.avg, m_acc.avg, results
