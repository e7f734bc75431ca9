#!/usr/bin/env python3
"""fp8 (e4m3) conv error on REAL vitals-range windows.

PARITY note #6 rejects fp8/MX conv on a dynamic-range argument; this
script turns the paragraph into numbers. For each precision treatment of
the MyCNN5 conv inputs+weights, it reports the error of (a) the conv1
output and (b) the end-to-end risk probability vs the fp32 reference, on:
  - real windows ETL'd from the committed p000194 record (raw vital
    ranges: HR~60-120 bpm, SpO2~95-100 %, RESP~10-30)
  - the same windows after per-window z-normalization (the opt-in
    `window_znorm` path — fp8's intended operating regime)

Treatments:
  bf16        — the shipped inference dtype (baseline for context)
  e4m3        — direct cast (torch.float8_e4m3fn), per-tensor no scaling
  e4m3+scale  — per-tensor absmax scaling to the e4m3 range
  e4m3 MX32   — per-32-element-block absmax scaling along the reduction
                axis (MXFP8-style, what a CDNA4 MX-MFMA kernel would use)

Quantization is simulated (quantize->dequantize, fp32 accumulate), which
UNDERSTATES real fp8 error if anything (real MFMA accumulates fp32 too,
so the sim matches the hardware path for these tiny reductions).

Usage: python scripts/fp8_error_study.py   (CPU or GPU)
"""
from __future__ import annotations

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402

E4M3_MAX = 448.0


def q_e4m3(t: torch.Tensor) -> torch.Tensor:
    return t.clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn).float()


def q_e4m3_scaled(t: torch.Tensor) -> torch.Tensor:
    s = t.abs().max().clamp_min(1e-12) / E4M3_MAX
    return q_e4m3(t / s) * s


def q_e4m3_mx32(t: torch.Tensor, dim: int) -> torch.Tensor:
    """Per-32-element-block absmax scaling along `dim` (MXFP8 geometry;
    real MX uses power-of-2 E8M0 scales — absmax fp32 scaling shown here
    is the OPTIMISTIC bound)."""
    n = t.shape[dim]
    pad = (32 - n % 32) % 32
    tt = t.movedim(dim, -1)
    if pad:
        tt = torch.cat([tt, torch.zeros(*tt.shape[:-1], pad)], dim=-1)
    blocks = tt.reshape(*tt.shape[:-1], -1, 32)
    s = blocks.abs().amax(dim=-1, keepdim=True).clamp_min(1e-12) / E4M3_MAX
    qb = q_e4m3(blocks / s) * s
    out = qb.reshape(*tt.shape)[..., :n]
    return out.movedim(-1, dim)


def study(x: np.ndarray, tag: str, model) -> list:
    xt = torch.from_numpy(x).float()
    age = torch.full((len(xt),), 65.0)
    w1 = model.conv1.weight.data.float()
    b1 = model.conv1.bias.data.float()
    ref_conv = torch.nn.functional.conv1d(xt, w1, b1)
    with torch.no_grad():
        ref_prob = torch.sigmoid(model(xt, age))

    def full_prob(xq, wq):
        import copy
        m = copy.deepcopy(model)
        m.conv1.weight.data = wq
        with torch.no_grad():
            return torch.sigmoid(m(xq, age))

    rows = []
    for name, (xq, wq) in {
        "bf16": (xt.bfloat16().float(), w1.bfloat16().float()),
        "e4m3": (q_e4m3(xt), q_e4m3(w1)),
        "e4m3+scale": (q_e4m3_scaled(xt), q_e4m3_scaled(w1)),
        "e4m3 MX32": (q_e4m3_mx32(xt, dim=2), q_e4m3_mx32(w1, dim=2)),
    }.items():
        c = torch.nn.functional.conv1d(xq, wq, b1)
        denom = ref_conv.abs().mean().clamp_min(1e-12)
        p = full_prob(xq, wq)
        rows.append({
            "windows": tag, "treatment": name,
            "conv_rel_err_mean": float((c - ref_conv).abs().mean() / denom),
            "conv_rel_err_max": float((c - ref_conv).abs().max() /
                                      ref_conv.abs().max().clamp_min(1e-12)),
            "prob_abs_err_mean": float((p - ref_prob).abs().mean()),
            "prob_abs_err_max": float((p - ref_prob).abs().max()),
        })
    return rows


def main() -> None:
    from tskd_amd.models import build_model
    from scripts.train_real_record import etl
    torch.manual_seed(0)
    model = build_model("MyCNN5").eval()
    # real checkpoint weights if present (trained this round on p000194)
    ck = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "artifacts", "MyCNN5_p000194.pth")
    if os.path.exists(ck):
        from tskd_amd.models import load_checkpoint
        model = load_checkpoint(ck).eval()
        print(f"[fp8] using trained checkpoint {ck}")
    x, ages, y, meta = etl()
    x = x[:256].astype(np.float32)
    rows = study(x, "raw vitals (p000194)", model)
    # z-normalized per window-row (the opt-in window_znorm path)
    m = x.mean(axis=2, keepdims=True)
    sd = x.std(axis=2, keepdims=True)
    xz = ((x - m) / np.maximum(sd, 1e-6)).astype(np.float32)
    rows += study(xz, "z-normalized", rows and model)
    for r in rows:
        print(json.dumps(r), flush=True)


if __name__ == "__main__":
    main()
