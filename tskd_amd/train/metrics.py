"""Training metrics — reference bin/utils.py:104-134 semantics."""

from __future__ import annotations

import torch


class AverageMeter:
    """Running average meter (reference utils.py:104-120)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val, n: int = 1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / max(self.count, 1)


def compute_batch_accuracy(output: torch.Tensor,
                           target: torch.Tensor) -> torch.Tensor:
    """sigmoid -> round -> eq -> mean (reference utils.py:122-134).
    `output` is logits, `target` in {0,1}."""
    with torch.no_grad():
        batch_size = target.size(0)
        pred = torch.sigmoid(output).round()
        correct = pred.eq(target.view_as(pred)).sum()
        return correct.float() * 100.0 / batch_size
