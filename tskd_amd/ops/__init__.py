"""HIP op dispatch for the MyCNN inference path.

GPU tensors run on the hand-written CDNA4 kernels (csrc/mycnn_kernels.hip)
— REQUIRED on a GPU box: if the extension .so is missing we raise instead of
silently falling back to eager PyTorch. CPU tensors use the PyTorch-eager
model itself (the numerics oracle the kernels are tested against).
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

from tskd_amd.ops import build as _build
from tskd_amd.ops.pack import (VARIANT_IDS, pack_conv1_mfma_bfrags,  # noqa: F401
                               pack_weights)

_lib: Optional[ctypes.CDLL] = None
_lib_err: Optional[str] = None


def _load_lib() -> ctypes.CDLL:
    global _lib, _lib_err
    if _lib is not None:
        return _lib
    path = _build.lib_path("_tskd_mycnn")
    if not os.path.exists(path):
        try:
            _build.build("_tskd_mycnn")
        except Exception as e:  # record, raise at use time
            _lib_err = f"hipcc build failed: {e}"
            raise RuntimeError(_lib_err)
    lib = ctypes.CDLL(path)
    lib.tskd_conv_fwd.restype = ctypes.c_int
    lib.tskd_conv_fwd.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p,
    ]
    lib.tskd_debug_mfma16x16x32.restype = ctypes.c_int
    lib.tskd_debug_mfma16x16x32.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
    ]
    lib.tskd_lstm_head_fwd.restype = ctypes.c_int
    lib.tskd_lstm_head_fwd.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int, ctypes.c_int, ctypes.c_float, ctypes.c_int,
        ctypes.c_int, ctypes.c_void_p,
    ]
    lib.tskd_pack_size.restype = ctypes.c_int
    lib.tskd_pack_size.argtypes = [ctypes.c_int]
    lib.tskd_feat_len.restype = ctypes.c_int
    lib.tskd_feat_len.argtypes = [ctypes.c_int]
    _lib = lib
    return lib


def hip_available() -> bool:
    try:
        _load_lib()
        return True
    except Exception:
        return False


def _stream_ptr() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


TLAST_SLACK = 256  # trailing elements the timelast conv kernel may overread


def alloc_windows(s: int, n: int, cin: int, win: int = 120,
                  timelast: bool = False, dtype=torch.bfloat16,
                  device="cuda") -> torch.Tensor:
    """Allocate a window tensor with the trailing slack the timelast
    (LDS-free) conv kernel requires; returns a (s, n, cin, win) or
    (s, n, win, cin) view tagged slack-safe."""
    numel = s * n * cin * win
    buf = torch.zeros(numel + TLAST_SLACK, dtype=dtype, device=device)
    shape = (s, n, win, cin) if timelast else (s, n, cin, win)
    t = buf[:numel].view(*shape)
    t._tskd_slack = True  # type: ignore[attr-defined]
    t._tskd_buf = buf  # keep storage alive  # type: ignore[attr-defined]
    return t


class MyCNNEngine:
    """Device-resident packed-weight inference engine for one MyCNN model.

    forward semantics == the reference model in eval mode,
    including the LSTM batch-axis-as-time quirk: ``x`` is (S, N, C, 120) —
    S independent sequences whose N windows each form one LSTM "batch".
    A reference-style single call is S=1.
    """

    def __init__(self, model, device: str = "cuda"):
        from tskd_amd.models.mycnn import _MyCNNBase
        assert isinstance(model, _MyCNNBase)
        self.variant = VARIANT_IDS[type(model).__name__]
        self.age_eps = float(model.AGE_EPS)
        self.cin = int(model.IN_CHANNELS)
        self.device = torch.device(device)
        self.model = model  # CPU oracle / fallback
        self.wpack = pack_weights(model).to(self.device)
        self.bfrag = pack_conv1_mfma_bfrags(model).to(self.device)
        if self.device.type == "cuda":
            lib = _load_lib()  # raises loudly if the HIP ext is unavailable
            expect = lib.tskd_pack_size(self.variant)
            if expect != self.wpack.numel():
                raise RuntimeError(
                    f"weight pack size mismatch: py={self.wpack.numel()} "
                    f"hip={expect}")
            self.feat_len = lib.tskd_feat_len(self.variant)
        else:
            self.feat_len = int(model.MAGICNUM)

    def conv_features(self, x: torch.Tensor) -> torch.Tensor:
        """(..., C, 120) or timelast (..., 120, C) -> (..., LIN) features.

        The timelast layout runs the LDS-free MFMA kernel (even channel
        counts, bf16); its fragment reads may overrun the tensor by up to
        TLAST_SLACK elements, so non-slack-tagged inputs are copied into a
        padded buffer (allocate with :func:`alloc_windows` to avoid that).
        """
        timelast = (x.shape[-1] == self.cin and x.shape[-2] == 120
                    and self.cin != 120)
        lead = x.shape[:-2]
        xf = x.reshape(-1, x.shape[-2], x.shape[-1]).contiguous()
        sn = xf.shape[0]
        if xf.device.type != "cuda":
            raise RuntimeError("conv_features is the GPU path; use the model on CPU")
        lib = _load_lib()
        if timelast and (self.cin % 2 != 0 or xf.dtype != torch.bfloat16):
            # odd-CIN / fp32: transpose back to the staged layout
            xf = xf.transpose(-1, -2).contiguous()
            timelast = False
        if xf.dtype == torch.bfloat16:
            is_bf16 = 1
        elif xf.dtype == torch.float32:
            is_bf16 = 0
        else:
            xf = xf.to(torch.float32)
            is_bf16 = 0
        if timelast and not getattr(x, "_tskd_slack", False):
            pad = alloc_windows(1, sn, self.cin, 120, timelast=True,
                                dtype=xf.dtype, device=xf.device)
            pad.reshape(-1).copy_(xf.reshape(-1))
            xf = pad.reshape(sn, 120, self.cin)
        feat = torch.empty(sn, self.feat_len, dtype=torch.float32, device=xf.device)
        rc = lib.tskd_conv_fwd(
            ctypes.c_void_p(xf.data_ptr()), is_bf16, int(timelast),
            ctypes.c_void_p(feat.data_ptr()),
            ctypes.c_void_p(self.wpack.data_ptr()),
            ctypes.c_void_p(self.bfrag.data_ptr()), sn, self.variant,
            _stream_ptr())
        if rc != 0:
            raise RuntimeError(f"tskd_conv_fwd failed: hipError {rc}")
        return feat.reshape(*lead, self.feat_len)

    def lstm_head(self, feat: torch.Tensor, age: Optional[torch.Tensor],
                  apply_sigmoid: bool = False) -> torch.Tensor:
        """(S, N, LIN) features -> (S, N) logits (or probabilities)."""
        assert feat.dim() == 3 and feat.shape[-1] == self.feat_len
        s, n = feat.shape[0], feat.shape[1]
        feat = feat.contiguous()
        lib = _load_lib()
        out = torch.empty(s, n, dtype=torch.float32, device=feat.device)
        age_ptr = ctypes.c_void_p(0)
        if age is not None:
            if age.dtype != torch.float32 or age.shape != (s, n) \
                    or not age.is_contiguous():
                age = age.to(torch.float32).expand(s, n).contiguous()
            age_ptr = ctypes.c_void_p(age.data_ptr())
        rc = lib.tskd_lstm_head_fwd(
            ctypes.c_void_p(feat.data_ptr()), age_ptr,
            ctypes.c_void_p(out.data_ptr()),
            ctypes.c_void_p(self.wpack.data_ptr()), s, n,
            ctypes.c_float(self.age_eps), int(apply_sigmoid), self.variant,
            _stream_ptr())
        if rc != 0:
            raise RuntimeError(f"tskd_lstm_head_fwd failed: hipError {rc}")
        return out

    def forward(self, x: torch.Tensor, age: Optional[torch.Tensor] = None,
                apply_sigmoid: bool = False) -> torch.Tensor:
        """x: (N, C, 120) single sequence, or (S, N, C, 120).

        Returns (N,) or (S, N) logits (probabilities if apply_sigmoid).
        """
        squeeze = x.dim() == 3
        if squeeze:
            x = x.unsqueeze(0)
            if age is not None and age.dim() == 1:
                age = age.unsqueeze(0)
        assert x.dim() == 4 and (
            (x.shape[-1] == 120 and x.shape[-2] == self.cin)
            or (x.shape[-2] == 120 and x.shape[-1] == self.cin))
        if x.device.type == "cuda":
            s, n = x.shape[0], x.shape[1]
            feat = self.conv_features(x).reshape(s, n, self.feat_len)
            out = self.lstm_head(feat, age, apply_sigmoid)
        else:
            s, n = x.shape[0], x.shape[1]
            if x.shape[-1] == self.cin and x.shape[-2] == 120:
                x = x.transpose(-1, -2)  # timelast input on the CPU path
            with torch.no_grad():
                outs = []
                for i in range(s):
                    a = age[i] if age is not None else torch.zeros(n)
                    y = self.model(x[i].float(), a.float())
                    outs.append(torch.sigmoid(y) if apply_sigmoid else y)
                out = torch.stack(outs)
        return out[0] if squeeze else out

    __call__ = forward


class GraphedForward:
    """hipGraph-captured micro-batch inference (BASELINE.json north star).

    Captures the fused conv+LSTM+head+sigmoid kernel sequence for a fixed
    (S, N) shape into a hipGraph (torch.cuda.CUDAGraph on ROCm) with static
    input/output buffers; replay eliminates per-kernel launch overhead in
    the serving hot loop. Fill ``x``/``age`` in place (or let the preprocess
    window-gather write directly into ``x``), then call ``replay()``.
    """

    def __init__(self, engine: "MyCNNEngine", s: int, n: int,
                 dtype: torch.dtype = torch.bfloat16,
                 apply_sigmoid: bool = True, warmup: int = 2,
                 timelast: bool = False, capture: bool = True):
        assert engine.device.type == "cuda"
        self.engine = engine
        self.timelast = timelast
        self.x = alloc_windows(s, n, engine.cin, 120, timelast=timelast,
                               dtype=dtype, device=engine.device)
        self.age = torch.full((s, n), 65.0, device=engine.device)
        self.graph = None
        if not capture:
            return  # buffers only (e.g. inside a TriggerGraph capture)
        stream = torch.cuda.Stream()
        with torch.cuda.stream(stream):
            for _ in range(warmup):
                out = engine.forward(self.x, self.age,
                                     apply_sigmoid=apply_sigmoid)
        torch.cuda.current_stream().wait_stream(stream)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out = engine.forward(self.x, self.age,
                                      apply_sigmoid=apply_sigmoid)

    def replay(self) -> torch.Tensor:
        self.graph.replay()
        return self.out
