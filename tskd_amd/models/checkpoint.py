"""Legacy-pickle checkpoint IO, bit-compatible with the reference .pth files.

The reference saves checkpoints as *legacy (non-zipfile) whole-module pickles*:
``torch.save(model, path, _use_new_zipfile_serialization=False)`` (reference
explore_torch.ipynb cell 26), producing a pickled ``__main__.MyCNN`` object
graph. Loading therefore requires a class importable as
``__main__.MyCNN`` (reference predictStream.py:8 imports it before :36 loads).

This module provides:
  - :func:`load_checkpoint` — loads any reference ``MyCNN*.pth`` (installing a
    ``__main__.MyCNN`` shim), detects the architecture variant from the
    restored conv1 shape, and re-classes the instance to the matching
    :mod:`tskd_amd.models.mycnn` variant so the correct forward runs.
  - :func:`save_checkpoint` — writes the SAME format (legacy pickle, global
    ``__main__.MyCNN``, FloatStorage) so the reference's
    ``torch.load(MODELPATH)`` round-trips our checkpoints unchanged.
"""

from __future__ import annotations

import sys
import warnings
from typing import Optional

import torch

from tskd_amd.models.mycnn import MyCNN2, MyCNN4, MyCNN5, _MyCNNBase


def _variant_for(conv1_weight: torch.Tensor) -> type:
    """Map restored conv1 shape -> architecture class."""
    shape = tuple(conv1_weight.shape)
    if shape == (4, 10, 10):
        return MyCNN5
    if shape == (4, 10, 5):
        return MyCNN4
    if shape == (4, 7, 5):
        return MyCNN2
    raise ValueError(f"unrecognized MyCNN conv1 shape {shape}")


class _MainShim:
    """Temporarily install a class as ``__main__.MyCNN`` (save & load both
    resolve the pickle GLOBAL through sys.modules['__main__'])."""

    def __init__(self, cls: type, name: str = "MyCNN"):
        self._cls, self._name = cls, name

    def __enter__(self):
        main = sys.modules["__main__"]
        self._had = hasattr(main, self._name)
        self._old = getattr(main, self._name, None)
        setattr(main, self._name, self._cls)
        return self

    def __exit__(self, *exc):
        main = sys.modules["__main__"]
        if self._had:
            setattr(main, self._name, self._old)
        else:
            delattr(main, self._name)
        return False


def load_checkpoint(path: str, map_location: str = "cpu") -> _MyCNNBase:
    """Load a legacy whole-module MyCNN pickle; returns an eval-mode module.

    Works on all four reference checkpoints (MyCNN2/3/4/5.pth) and on files
    written by :func:`save_checkpoint`.
    """
    with _MainShim(MyCNN5):
        with warnings.catch_warnings():
            # Legacy container-source warnings are expected: the embedded
            # source is the reference's, ours differs by design.
            warnings.simplefilter("ignore")
            model = torch.load(path, map_location=map_location, weights_only=False)
    if not isinstance(model, torch.nn.Module):
        raise TypeError(f"{path} did not contain an nn.Module (got {type(model)})")
    # Re-class to the matching variant so the right forward/constants apply.
    cls = _variant_for(model.conv1.weight.data)
    model.__class__ = cls
    model.eval()
    return model


def save_checkpoint(model: _MyCNNBase, path: str, main_compat: bool = True) -> None:
    """Write a reference-format checkpoint (legacy pickle, ``__main__.MyCNN``).

    With ``main_compat`` the pickled GLOBAL is ``__main__.MyCNN`` exactly as
    the reference writes it, so ``predictStream.py`` in the reference repo
    loads our files without modification.
    """
    orig_cls: Optional[type] = None
    if main_compat:
        # Pickle records the class by __module__/__qualname__ and verifies the
        # lookup; a dynamic subclass registered in __main__ satisfies both.
        orig_cls = model.__class__
        shim = type("MyCNN", (orig_cls,), {})
        shim.__module__ = "__main__"
        shim.__qualname__ = "MyCNN"
        model.__class__ = shim
        ctx: object = _MainShim(shim)
    else:
        class _Null:
            def __enter__(self):
                return self

            def __exit__(self, *e):
                return False
        ctx = _Null()
    try:
        with ctx, warnings.catch_warnings():
            warnings.simplefilter("ignore")
            torch.save(model, path, _use_new_zipfile_serialization=False)
    finally:
        if orig_cls is not None:
            model.__class__ = orig_cls
