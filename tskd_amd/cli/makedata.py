"""Synthetic stream CSV generator — reference data/makeData.py semantics:
one value per second for 24 h, random ints 0-100, ~50% row dropout,
deterministic (seed 42)."""

from __future__ import annotations

import argparse

import numpy as np


def make_data(path: str = "data.csv", seed: int = 42, hours: float = 24.0,
              dropout: float = 0.5) -> int:
    np.random.seed(seed)
    n = int(hours * 3600)
    ts = np.arange(n)
    vals = np.random.randint(0, 101, size=n)
    keep = np.random.random(n) >= dropout
    with open(path, "w") as f:
        f.write("timestamp,value\n")
        for t, v in zip(ts[keep], vals[keep]):
            f.write(f"{t},{v}\n")
    return int(keep.sum())


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--out", default="data.csv")
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--hours", type=float, default=24.0)
    ap.add_argument("--dropout", type=float, default=0.5)
    args = ap.parse_args(argv)
    n = make_data(args.out, args.seed, args.hours, args.dropout)
    print(f"wrote {n} rows to {args.out}")


if __name__ == "__main__":
    main()
