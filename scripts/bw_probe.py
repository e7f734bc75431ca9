#!/usr/bin/env python3
"""Run the HBM read-ceiling probe (scripts/bwprobe.hip) on an MI355X.

Prints TB/s for each (load-ILP, nontemporal, grid) combination over an 8 GB
float buffer, giving the calibration number the ingest stage is priced
against in profiles/r01_kernel_profiles.md.
"""

import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

LIB = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_bwprobe.so")


def main() -> None:
    assert torch.cuda.is_available()
    lib = ctypes.CDLL(LIB)
    lib.bw_probe.restype = ctypes.c_int
    lib.bw_probe.argtypes = [ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p,
                             ctypes.c_int, ctypes.c_int, ctypes.c_void_p]
    n = 2 * 1024 * 1024 * 1024  # floats -> 8 GiB
    x = torch.randn(n, device="cuda")
    out = torch.empty(65536, device="cuda")
    names = ["x4 nt", "x8 nt", "x16 nt", "x4", "x8", "x16",
         "grouped", "contig", "consume", "packed"]
    stream = torch.cuda.current_stream().cuda_stream
    best = (0.0, "")
    for variant in range(10):
        for grid in (2048, 4096, 8192, 16384, 32768):
            rc = lib.bw_probe(x.data_ptr(), n, out.data_ptr(), variant,
                              grid, stream)
            assert rc == 0, rc
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            reps = 6
            for _ in range(reps):
                lib.bw_probe(x.data_ptr(), n, out.data_ptr(), variant,
                             grid, stream)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / reps
            tbs = n * 4 / dt / 1e12
            tag = f"{names[variant]:7s} grid={grid:6d}"
            print(f"{tag}  {tbs:6.2f} TB/s")
            if tbs > best[0]:
                best = (tbs, tag)
    print(f"BEST: {best[1]}  {best[0]:.2f} TB/s")


if __name__ == "__main__":
    main()
