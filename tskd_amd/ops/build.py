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


# Host-side C++ pybind11 extensions (bus, store, wfdb — no GPU code).
PKG_DIR = os.path.dirname(OPS_DIR)
HOST_CSRC = os.path.join(PKG_DIR, "csrc")
HOST_MODULES = {
    "_tskd_bus": "bus.cpp",
    "_tskd_store": "store.cpp",
    "_tskd_wfdb": "wfdb.cpp",
}


def _ext_suffix() -> str:
    import sysconfig
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def host_lib_path(name: str) -> str:
    return os.path.join(PKG_DIR, name + _ext_suffix())


def build_host(name: str, force: bool = False, verbose: bool = True) -> str:
    import pybind11
    import sysconfig
    src = os.path.join(HOST_CSRC, HOST_MODULES[name])
    out = host_lib_path(name)
    if not force and os.path.exists(out) and \
            os.path.getmtime(out) >= os.path.getmtime(src):
        return out
    cmd = [
        "g++", "-O2", "-std=c++17", "-shared", "-fPIC",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        src, "-o", out, "-lpthread",
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
    for name, src in HOST_MODULES.items():
        if os.path.exists(os.path.join(HOST_CSRC, src)):
            outs.append(build_host(name, force=force))
    return outs


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
