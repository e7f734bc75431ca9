"""In-tree build of the HIP kernel library for gfx950 (MI355X).

hipcc cross-compiles without a GPU; the produced .so is committed-adjacent
(git-ignored) and ships to the GPU box with the repo snapshot.
"""

from __future__ import annotations

import os
import subprocess
import sys

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
LIB_DIR = os.path.join(OPS_DIR, "lib")
ARCH = os.environ.get("TSKD_GPU_ARCH", "gfx950")

SOURCES = {
    "_tskd_mycnn": ["mycnn_kernels.hip"],
    "_tskd_preprocess": ["preprocess_kernels.hip"],
    "_tskd_train": ["train_kernels.hip"],
}


def lib_path(name: str) -> str:
    return os.path.join(LIB_DIR, f"{name}_{ARCH}.so")


def _hipcc() -> str:
    for cand in (os.environ.get("HIPCC"), "/opt/rocm/bin/hipcc", "hipcc"):
        if cand and (os.path.exists(cand) or cand == "hipcc"):
            return cand
    return "hipcc"


def build(name: str, force: bool = False, verbose: bool = True) -> str:
    srcs = [os.path.join(CSRC, s) for s in SOURCES[name]]
    missing = [s for s in srcs if not os.path.exists(s)]
    if missing:
        raise FileNotFoundError(missing)
    out = lib_path(name)
    os.makedirs(LIB_DIR, exist_ok=True)
    if not force and os.path.exists(out) and all(
        os.path.getmtime(out) >= os.path.getmtime(s) for s in srcs
    ):
        return out
    cmd = [
        _hipcc(), f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", *srcs, "-o", out,
    ]
    if verbose:
        print("[tskd build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return out


def build_all(force: bool = False) -> list:
    outs = []
    for name, srcs in SOURCES.items():
        if all(os.path.exists(os.path.join(CSRC, s)) for s in srcs):
            outs.append(build(name, force=force))
    return outs


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
