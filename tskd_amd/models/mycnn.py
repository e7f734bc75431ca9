"""MyCNN model family — CPU-truth definitions (PyTorch eager).

These are the *reference-semantics* definitions used as the numerics oracle
for the HIP kernels in :mod:`tskd_amd.ops` and as the class targets for
legacy-pickle checkpoint IO (:mod:`tskd_amd.models.checkpoint`).

Architecture sources (cited for parity checking, code written from scratch):
  - MyCNN5 (the served model): reference bin/models.py:5-36 —
    Conv1d(10,4,k=10) -> tanh -> MaxPool1d(3,2) -> Dropout(0.1) ->
    Conv1d(4,1,k=5) -> tanh -> pool -> dropout -> view(-1,25) ->
    LSTM(25,16,2) -> Linear(16,1) -> * relu(age*1e-8 + 1) -> squeeze.
    Dead weights carried for checkpoint compatibility: out1 Linear(567,1),
    out2 Linear(16,1), age_fn Linear(1,1) (never called in forward).
  - MyCNN2/MyCNN3 (7-channel) and MyCNN4 (10-channel): reference
    "explore_torch copy.ipynb" cell 1 — Conv1d(C,4,k=5) -> tanh ->
    MaxPool1d(2,2) -> Conv1d(4,1,k=5) -> tanh -> pool -> Dropout(0.5)
    (single dropout site) -> view(-1,27) -> LSTM(27,16,2) -> Linear(16,1)
    -> * relu(age*1e-4 + 1). Channel counts per checkpoint state shapes
    (MyCNN2/3: conv1 (4,7,5); MyCNN4: conv1 (4,10,5)) — see SURVEY.md §2.3.

THE LSTM-OVER-BATCH QUIRK (load-bearing for bit-compatibility): the LSTM is
called on a 2-D tensor ``(N, feat)``, which PyTorch treats as an *unbatched*
sequence — the batch axis becomes the TIME axis, so hidden state flows across
windows within a batch and the output of window ``i`` depends on windows
``0..i-1``. Every numerics test and every HIP kernel must reproduce this
(reference bin/models.py:30).

``sigmoid`` is applied by the CALLER, not by forward (reference utils.py:261,
predictStream.py:160).
"""

from __future__ import annotations

import torch
import torch.nn as nn


class _MyCNNBase(nn.Module):
    """Shared structure. Subclasses set the conv/pool/LSTM geometry."""

    IN_CHANNELS: int = 10
    CONV1_K: int = 10
    POOL_K: int = 3
    POOL_S: int = 2
    DROPOUT_P: float = 0.1
    LSTM_IN: int = 25
    AGE_EPS: float = 1e-8
    TWO_DROPOUT_SITES: bool = True  # MyCNN5 has dropout after each pool

    def __init__(self) -> None:
        self.MAGICNUM = self.LSTM_IN
        super().__init__()
        self.conv1 = nn.Conv1d(self.IN_CHANNELS, 4, kernel_size=self.CONV1_K)
        self.conv2 = nn.Conv1d(4, 1, kernel_size=5)
        self.pool = nn.MaxPool1d(kernel_size=self.POOL_K, stride=self.POOL_S)
        self.out1 = nn.Linear(567, 1)  # dead weight, kept for ckpt parity
        self.dropout = nn.Dropout(self.DROPOUT_P)
        self.lstm = nn.LSTM(input_size=self.LSTM_IN, hidden_size=16, num_layers=2)
        self.out = nn.Linear(16, 1, bias=True)
        self.out2 = nn.Linear(16, 1, bias=True)  # dead weight
        self.age_fn = nn.Linear(1, 1, bias=True)  # dead weight in MyCNN5 forward

    def forward(self, x: torch.Tensor, age: torch.Tensor) -> torch.Tensor:
        x = torch.tanh(self.conv1(x))
        x = self.pool(x)
        if self.TWO_DROPOUT_SITES:
            x = self.dropout(x)
        x = torch.tanh(self.conv2(x))
        x = self.pool(x)
        x = self.dropout(x)
        x = x.view(-1, self.MAGICNUM)
        # 2-D input => unbatched: batch axis IS the sequence axis.
        x, _ = self.lstm(x)
        x = self.out(x)
        age_scale = torch.relu(age.unsqueeze(1) * self.AGE_EPS + 1)
        x = x * age_scale
        return x.squeeze(1)


class MyCNN5(_MyCNNBase):
    """The served model (reference config.cfg:9 MODELPATH -> MyCNN5.pth)."""


# ``MyCNN`` is the class name embedded in every reference checkpoint pickle;
# alias it to the served architecture like reference bin/models.py does.
MyCNN = MyCNN5


class MyCNN4(_MyCNNBase):
    """10-channel k=5 variant (ckpt conv1 (4,10,5))."""

    CONV1_K = 5
    POOL_K = 2
    DROPOUT_P = 0.5
    LSTM_IN = 27
    AGE_EPS = 1e-4
    TWO_DROPOUT_SITES = False


class MyCNN2(MyCNN4):
    """7-channel variant (ckpt conv1 (4,7,5))."""

    IN_CHANNELS = 7


class MyCNN3(MyCNN2):
    """Same architecture as MyCNN2, different training run."""


_VARIANTS = {"MyCNN": MyCNN5, "MyCNN2": MyCNN2, "MyCNN3": MyCNN3,
             "MyCNN4": MyCNN4, "MyCNN5": MyCNN5}


def build_model(name: str = "MyCNN5") -> _MyCNNBase:
    try:
        return _VARIANTS[name]()
    except KeyError:
        raise ValueError(f"unknown model variant {name!r}; have {sorted(_VARIANTS)}")
