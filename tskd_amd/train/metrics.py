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
    `output` is logits, `target` in {0,1}. On CUDA this is the fused K14
    metric kernel (one pass, no intermediate tensors); the HIP extension
    is REQUIRED there — no silent eager fallback."""
    with torch.no_grad():
        batch_size = target.size(0)
        if output.is_cuda:
            return _batch_accuracy_hip(output, target, batch_size)
        pred = torch.sigmoid(output).round()
        correct = pred.eq(target.view_as(pred)).sum()
        return correct.float() * 100.0 / batch_size


def _batch_accuracy_hip(output: torch.Tensor, target: torch.Tensor,
                        batch_size: int) -> torch.Tensor:
    """K14 fused metric kernel (ops/csrc/train_kernels.hip)."""
    import ctypes

    from tskd_amd.train.hip_trainer import _load_train_lib, _sp
    lib = _load_train_lib()
    lo = output.detach().reshape(-1).float().contiguous()
    tg = target.detach().reshape(-1).float().contiguous()
    out = torch.zeros(1, device=output.device)
    rc = lib.tskd_train_batch_accuracy(
        ctypes.c_void_p(lo.data_ptr()), ctypes.c_void_p(tg.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), ctypes.c_long(lo.numel()), _sp())
    if rc != 0:
        raise RuntimeError(f"batch_accuracy kernel failed: hipError {rc}")
    return out[0] * 100.0 / batch_size
