"""Weight packing for the HIP kernels.

Layout MUST match ``Geom`` in csrc/mycnn_kernels.hip:
  [ W1(4,CIN,K1) | b1(4) | W2(4,5) | b2(1)
  | l1.W_ih(64,LIN) | l1.W_hh(64,16) | l1.bias(64)=b_ih+b_hh
  | l2.W_ih(64,16)  | l2.W_hh(64,16) | l2.bias(64)
  | out.W(16) | out.b(1) ]
PyTorch LSTM gate row order (i,f,g,o) is kept as-is — the kernel's lane
layout assumes it.
"""

from __future__ import annotations

import torch

VARIANT_IDS = {"MyCNN5": 0, "MyCNN": 0, "MyCNN2": 1, "MyCNN3": 1, "MyCNN4": 2}


def pack_weights(model) -> torch.Tensor:
    sd = {k: v.detach().float().cpu() for k, v in model.state_dict().items()}
    parts = [
        sd["conv1.weight"].reshape(-1),
        sd["conv1.bias"].reshape(-1),
        sd["conv2.weight"].reshape(-1),
        sd["conv2.bias"].reshape(-1),
        sd["lstm.weight_ih_l0"].reshape(-1),
        sd["lstm.weight_hh_l0"].reshape(-1),
        (sd["lstm.bias_ih_l0"] + sd["lstm.bias_hh_l0"]).reshape(-1),
        sd["lstm.weight_ih_l1"].reshape(-1),
        sd["lstm.weight_hh_l1"].reshape(-1),
        (sd["lstm.bias_ih_l1"] + sd["lstm.bias_hh_l1"]).reshape(-1),
        sd["out.weight"].reshape(-1),
        sd["out.bias"].reshape(-1),
    ]
    return torch.cat(parts).contiguous()
