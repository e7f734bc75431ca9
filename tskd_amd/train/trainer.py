"""High-level trainer — reference explore_torch.ipynb cell 26 recipe:
BCEWithLogitsLoss(pos_weight = n_neg/n_pos), Adam(lr=1e-5), shuffled
DataLoader, best-validation-loss checkpoint in the legacy pickle format."""

from __future__ import annotations

import copy
from typing import Optional, Tuple

import numpy as np
import torch
from torch.utils.data import DataLoader

from tskd_amd.models import build_model, save_checkpoint
from tskd_amd.train.data import load_dataset
from tskd_amd.train.loop import evaluate, train


def fit(x_train: np.ndarray, age_train: np.ndarray, y_train: np.ndarray,
        x_val: np.ndarray, age_val: np.ndarray, y_val: np.ndarray,
        variant: str = "MyCNN5", epochs: int = 250, batch_size: int = 64,
        lr: float = 1e-5, device: str = "cpu",
        checkpoint_path: Optional[str] = None, print_freq: int = 100,
        seed: int = 0) -> Tuple[torch.nn.Module, dict]:
    torch.manual_seed(seed)
    model = build_model(variant).to(device)
    n_pos = max(float((y_train == 1).sum()), 1.0)
    n_neg = float((y_train == 0).sum())
    criterion = torch.nn.BCEWithLogitsLoss(
        pos_weight=torch.tensor(n_neg / n_pos, device=device))
    optimizer = torch.optim.Adam(model.parameters(), lr=lr)
    train_loader = DataLoader(load_dataset(x_train, age_train, y_train),
                              batch_size=batch_size, shuffle=True,
                              generator=torch.Generator().manual_seed(seed))
    val_loader = DataLoader(load_dataset(x_val, age_val, y_val),
                            batch_size=batch_size)
    best_val = float("inf")
    best_state = None
    history = {"train_loss": [], "val_loss": [], "val_acc": []}
    for epoch in range(epochs):
        tl, _ = train(model, device, train_loader, criterion, optimizer,
                      epoch, print_freq=print_freq)
        vl, va, _ = evaluate(model, device, val_loader, criterion,
                             print_freq=print_freq)
        history["train_loss"].append(tl)
        history["val_loss"].append(vl)
        history["val_acc"].append(va)
        if vl < best_val:
            best_val = vl
            best_state = copy.deepcopy(model.state_dict())
            if checkpoint_path:
                save_checkpoint(model.cpu().eval(), checkpoint_path)
                model.to(device).train()
    if best_state is not None:
        model.load_state_dict(best_state)
    model.eval()
    history["best_val_loss"] = best_val
    return model, history
