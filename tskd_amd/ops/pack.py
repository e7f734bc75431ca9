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


def pack_conv1_mfma_bfrags(model) -> torch.Tensor:
    """Pack conv1 weights into v_mfma_f32_16x16x32_bf16 B-fragment order.

    Returns bf16 tensor [KSTEPS, 64, 8]: lane l of k-step st holds
    B[k = (l>>4)*8 + j][col = l&15] where B[kk][c] = W1[c][kk % CIN][kk // CIN]
    (the im2col reduction index kk = k*CIN + i matching the kernel's
    transposed-window A view), zero-padded for kk >= CIN*K1 and col >= 4.
    """
    w1 = model.conv1.weight.detach().float().cpu()  # (4, CIN, K1)
    _, cin, k1 = w1.shape
    kk_n = cin * k1
    ksteps = (kk_n + 31) // 32
    out = torch.zeros(ksteps, 64, 8, dtype=torch.float32)
    for st in range(ksteps):
        for lane in range(64):
            col = lane & 15
            if col >= 4:
                continue
            for j in range(8):
                kk = st * 32 + (lane >> 4) * 8 + j
                if kk < kk_n:
                    out[st, lane, j] = w1[col, kk % cin, kk // cin]
    return out.to(torch.bfloat16).contiguous()


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


def pack_offsets(model) -> dict:
    """Python mirror of the Geom<> pack offsets (see mycnn_kernels.hip)."""
    cin = int(model.IN_CHANNELS)
    k1 = int(model.CONV1_K)
    lin = int(model.LSTM_IN)
    o = {}
    o["w1"] = (0, 4 * cin * k1)
    o["b1"] = (o["w1"][1], o["w1"][1] + 4)
    o["w2"] = (o["b1"][1], o["b1"][1] + 20)
    o["b2"] = (o["w2"][1], o["w2"][1] + 1)
    o["wih1"] = (o["b2"][1], o["b2"][1] + 64 * lin)
    o["whh1"] = (o["wih1"][1], o["wih1"][1] + 64 * 16)
    o["bl1"] = (o["whh1"][1], o["whh1"][1] + 64)
    o["wih2"] = (o["bl1"][1], o["bl1"][1] + 64 * 16)
    o["whh2"] = (o["wih2"][1], o["wih2"][1] + 64 * 16)
    o["bl2"] = (o["whh2"][1], o["whh2"][1] + 64)
    o["outw"] = (o["bl2"][1], o["bl2"][1] + 16)
    o["outb"] = (o["outw"][1], o["outw"][1] + 1)
    o["npack"] = o["outb"][1]
    return o


def unpack_weights_into(model, wpack) -> None:
    """Write a packed parameter vector back into a MyCNN module.

    The combined LSTM bias is split half/half between bias_ih and bias_hh
    (only the sum enters the forward math — documented parameterization
    choice of the packed trainer)."""
    import torch as _t
    o = pack_offsets(model)
    w = wpack.detach().float().cpu()

    def sl(name):
        a, b = o[name]
        return w[a:b]

    cin, k1, lin = int(model.IN_CHANNELS), int(model.CONV1_K), int(model.LSTM_IN)
    with _t.no_grad():
        model.conv1.weight.copy_(sl("w1").view(4, cin, k1))
        model.conv1.bias.copy_(sl("b1"))
        model.conv2.weight.copy_(sl("w2").view(1, 4, 5))
        model.conv2.bias.copy_(sl("b2"))
        model.lstm.weight_ih_l0.copy_(sl("wih1").view(64, lin))
        model.lstm.weight_hh_l0.copy_(sl("whh1").view(64, 16))
        model.lstm.bias_ih_l0.copy_(sl("bl1") * 0.5)
        model.lstm.bias_hh_l0.copy_(sl("bl1") * 0.5)
        model.lstm.weight_ih_l1.copy_(sl("wih2").view(64, 16))
        model.lstm.weight_hh_l1.copy_(sl("whh2").view(64, 16))
        model.lstm.bias_ih_l1.copy_(sl("bl2") * 0.5)
        model.lstm.bias_hh_l1.copy_(sl("bl2") * 0.5)
        model.out.weight.copy_(sl("outw").view(1, 16))
        model.out.bias.copy_(sl("outb"))
