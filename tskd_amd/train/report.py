"""Evaluation reports — reference explore_torch.ipynb cells 27-46:
classification report, ROC-AUC, per-patient F1 tables, and the sklearn
LogisticRegression / SVC sanity baselines on flattened windows."""

from __future__ import annotations

from typing import Dict, Optional, Sequence

import numpy as np
import torch


def score_model(model, x: np.ndarray, age: np.ndarray,
                batch_size: int = 256) -> np.ndarray:
    """Probabilities for (n, C, 120) windows (CPU eval path)."""
    model.eval()
    probs = []
    with torch.no_grad():
        for i in range(0, len(x), batch_size):
            xb = torch.from_numpy(x[i:i + batch_size]).float()
            ab = torch.from_numpy(age[i:i + batch_size]).float()
            probs.append(torch.sigmoid(model(xb, ab)).numpy())
    return np.concatenate(probs) if probs else np.empty(0)


def classification_metrics(y_true: np.ndarray, y_prob: np.ndarray,
                           threshold: float = 0.5) -> Dict:
    """ROC-AUC + precision/recall/F1 (cells 31-35)."""
    from sklearn.metrics import (classification_report, f1_score,
                                 roc_auc_score)
    y_pred = (y_prob >= threshold).astype(int)
    out = {
        "n": int(len(y_true)),
        "pos_rate": float(np.mean(y_true)),
        "f1": float(f1_score(y_true, y_pred, zero_division=0)),
        "report": classification_report(y_true, y_pred, zero_division=0),
    }
    try:
        out["roc_auc"] = float(roc_auc_score(y_true, y_prob))
    except ValueError:
        out["roc_auc"] = float("nan")
    return out


def per_patient_f1(y_true: np.ndarray, y_prob: np.ndarray,
                   patient_ids: Sequence[str],
                   threshold: float = 0.5) -> Dict[str, float]:
    """Per-patient F1 table (cell 35)."""
    from sklearn.metrics import f1_score
    pids = np.asarray(patient_ids)
    out = {}
    for pid in np.unique(pids):
        m = pids == pid
        out[str(pid)] = float(f1_score(y_true[m],
                                       (y_prob[m] >= threshold).astype(int),
                                       zero_division=0))
    return out


def sklearn_baselines(x: np.ndarray, y: np.ndarray,
                      x_test: Optional[np.ndarray] = None,
                      y_test: Optional[np.ndarray] = None,
                      svc: bool = False) -> Dict[str, Dict]:
    """LogisticRegression (+ optional SVC) on flattened (n, C*120) windows —
    the reference's is-there-signal sanity check (cells 44-46)."""
    from sklearn.linear_model import LogisticRegression
    from sklearn.metrics import roc_auc_score
    xf = x.reshape(len(x), -1)
    xt = x_test.reshape(len(x_test), -1) if x_test is not None else xf
    yt = y_test if y_test is not None else y
    out = {}
    lr = LogisticRegression(max_iter=200)
    lr.fit(xf, y)
    out["logreg"] = {
        "train_acc": float(lr.score(xf, y)),
        "test_acc": float(lr.score(xt, yt)),
        "test_auc": float(roc_auc_score(yt, lr.predict_proba(xt)[:, 1])),
    }
    if svc:
        from sklearn.svm import SVC
        sv = SVC(probability=False)
        sv.fit(xf, y)
        out["svc"] = {"train_acc": float(sv.score(xf, y)),
                      "test_acc": float(sv.score(xt, yt))}
    return out
