"""MyCNNHipTrainer — the MI355X-native training step (BASELINE config 5).

Runs the full training step on the hand-written HIP kernel set
(csrc/train_kernels.hip): fused conv forward-with-stash, LSTM scan
forward-with-stash, BCEWithLogits(pos_weight) loss+grad, reverse BPTT and
conv backward, DP gradient all-reduce (RCCL over xGMI when distributed),
and a fused Adam step over the single packed parameter buffer.

Semantics match the reference training recipe (explore_torch.ipynb cell 26;
tskd_amd/train/trainer.py is the torch-eager parity implementation): the
LSTM batch-axis-as-time quirk holds per sequence, targets are per-window
{0,1}, loss is the mean weighted BCE over all S*B windows.
"""

from __future__ import annotations

import ctypes
from typing import Optional, Tuple

import torch

from tskd_amd.ops import build as _build
from tskd_amd.ops.pack import (VARIANT_IDS, pack_offsets, pack_weights,
                               unpack_weights_into)

_tlib: Optional[ctypes.CDLL] = None


def _load_train_lib() -> ctypes.CDLL:
    global _tlib
    if _tlib is not None:
        return _tlib
    import os
    path = _build.lib_path("_tskd_train")
    if not os.path.exists(path):
        _build.build("_tskd_train")
    lib = ctypes.CDLL(path)
    P, I, L, F = ctypes.c_void_p, ctypes.c_int, ctypes.c_long, ctypes.c_float
    lib.tskd_train_conv_fwd.restype = I
    lib.tskd_train_conv_fwd.argtypes = [P, P, P, P, I, F, F,
                                        ctypes.c_ulonglong, I, P]
    lib.tskd_train_lstm_fwd.restype = I
    lib.tskd_train_lstm_fwd.argtypes = [P, P, P, P, P, I, I, F, I, P]
    lib.tskd_train_loss.restype = I
    lib.tskd_train_loss.argtypes = [P, P, P, P, P, L, F, F, P]
    lib.tskd_train_lstm_bwd.restype = I
    lib.tskd_train_lstm_bwd.argtypes = [P, P, P, P, P, P, I, I, I, P]
    lib.tskd_train_conv_bwd.restype = I
    lib.tskd_train_conv_bwd.argtypes = [P, P, P, P, P, I, I, P]
    lib.tskd_train_adam.restype = I
    lib.tskd_train_adam.argtypes = [P, P, P, P, L, F, F, F, F, I, P]
    lib.tskd_train_stash_sizes.restype = I
    lib.tskd_train_stash_sizes.argtypes = [I, P, P]
    lib.tskd_train_batch_accuracy.restype = I
    lib.tskd_train_batch_accuracy.argtypes = [P, P, P, L, P]
    _tlib = lib
    return lib


def _sp() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _p(t: torch.Tensor) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr())


class MyCNNHipTrainer:
    def __init__(self, model, device: str = "cuda", lr: float = 1e-5,
                 betas: Tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                 pos_weight: float = 1.0, train_dropout: bool = True,
                 seed: int = 0):
        self.model = model
        # reference train-mode dropout sites: MyCNN5 drops
        # after BOTH pools; MyCNN2/3/4 only after pool2
        p_drop = float(model.DROPOUT_P) if train_dropout else 0.0
        self.drop1_p = p_drop if getattr(model, "TWO_DROPOUT_SITES", True) \
            else 0.0
        self.drop2_p = p_drop
        self.seed = int(seed)
        self.variant = VARIANT_IDS[type(model).__name__]
        self.age_eps = float(model.AGE_EPS)
        self.cin = int(model.IN_CHANNELS)
        self.lin = int(model.MAGICNUM)
        self.device = torch.device(device)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.pos_weight = float(pos_weight)
        self.offsets = pack_offsets(model)
        self.wpack = pack_weights(model).to(self.device)
        self.grads = torch.zeros_like(self.wpack)
        self.m = torch.zeros_like(self.wpack)
        self.v = torch.zeros_like(self.wpack)
        self.step_count = 0
        self._probe_stash_sizes()

    def _probe_stash_sizes(self):
        lib = _load_train_lib()
        cw, lw = ctypes.c_int(), ctypes.c_int()
        rc = lib.tskd_train_stash_sizes(self.variant, ctypes.byref(cw),
                                        ctypes.byref(lw))
        assert rc == 0
        self._stash_conv_words = cw.value
        self._stash_lstm_words = lw.value

    def forward_backward(self, x: torch.Tensor, age: Optional[torch.Tensor],
                         y: torch.Tensor) -> float:
        """x (S, B, C, 120) fp32, y (S, B) in {0,1}. Accumulates into grads;
        returns the scalar loss."""
        lib = _load_train_lib()
        assert x.dim() == 4 and x.shape[2] == self.cin and x.shape[3] == 120
        S, B = x.shape[0], x.shape[1]
        SN = S * B
        dev = self.device
        x = x.to(dev, torch.float32).contiguous()
        y = y.to(dev, torch.float32).reshape(S, B).contiguous()
        if age is not None:
            age = age.to(dev, torch.float32).expand(S, B).contiguous()
        feat = torch.empty(SN, self.lin, device=dev)
        stash_c = torch.empty(SN, self._stash_conv_words, device=dev)
        self._last_stash_conv = stash_c  # exposed for mask-exact testing
        stash_l = torch.empty(S, B, self._stash_lstm_words, device=dev)
        logits = torch.empty(S, B, device=dev)
        dlogit = torch.empty(S, B, device=dev)
        dfeat = torch.empty(S, B, self.lin, device=dev)
        loss = torch.zeros(1, device=dev)
        st = _sp()
        age_p = _p(age) if age is not None else ctypes.c_void_p(0)
        self.seed += 1  # fresh dropout masks every call
        rc = lib.tskd_train_conv_fwd(
            _p(x), _p(feat), _p(stash_c), _p(self.wpack), SN,
            ctypes.c_float(self.drop1_p), ctypes.c_float(self.drop2_p),
            ctypes.c_ulonglong(self.seed), self.variant, st)
        assert rc == 0, f"conv_fwd {rc}"
        rc = lib.tskd_train_lstm_fwd(_p(feat), age_p, _p(self.wpack),
                                     _p(stash_l), _p(logits), S, B,
                                     self.age_eps, self.variant, st)
        assert rc == 0, f"lstm_fwd {rc}"
        rc = lib.tskd_train_loss(_p(logits), _p(y), age_p, _p(dlogit),
                                 _p(loss), SN, self.pos_weight, self.age_eps,
                                 st)
        assert rc == 0, f"loss {rc}"
        rc = lib.tskd_train_lstm_bwd(_p(feat), _p(dlogit), _p(stash_l),
                                     _p(self.wpack), _p(self.grads),
                                     _p(dfeat), S, B, self.variant, st)
        assert rc == 0, f"lstm_bwd {rc}"
        rc = lib.tskd_train_conv_bwd(_p(x), _p(stash_c), _p(dfeat),
                                     _p(self.wpack), _p(self.grads), SN,
                                     self.variant, st)
        assert rc == 0, f"conv_bwd {rc}"
        return float(loss.item())

    def optimizer_step(self) -> None:
        """DP all-reduce (when distributed) + fused Adam + grad zeroing."""
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(self.grads)
            self.grads /= dist.get_world_size()
        self.step_count += 1
        lib = _load_train_lib()
        rc = lib.tskd_train_adam(_p(self.wpack), _p(self.grads), _p(self.m),
                                 _p(self.v), self.wpack.numel(), self.lr,
                                 self.betas[0], self.betas[1], self.eps,
                                 self.step_count, _sp())
        assert rc == 0, f"adam {rc}"

    def step(self, x, age, y) -> float:
        loss = self.forward_backward(x, age, y)
        assert loss == loss, "model diverged with loss = NaN"  # tripwire
        self.optimizer_step()
        return loss

    def export_model(self):
        unpack_weights_into(self.model, self.wpack)
        return self.model
