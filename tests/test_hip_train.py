"""HIP training-path numerics vs torch autograd (the fp32 oracle).

Gradient parity covers the whole K10-K12 chain: BCE-with-logits(pos_weight)
-> head + age gate -> reverse BPTT through the 2-layer batch-as-time LSTM ->
maxpool/tanh/conv backward. The oracle maps torch autograd grads onto the
packed layout (d bias_ih == d bias_hh == d combined bias).
"""

import numpy as np
import pytest
import torch

from tskd_amd.models import build_model
from tskd_amd.ops.pack import pack_offsets, pack_weights


def _torch_loss_and_grads(model, x, age, y, pos_weight):
    """Oracle: autograd over the per-sequence forward; packed-layout grads."""
    model.zero_grad()
    logits = []
    for s in range(x.shape[0]):
        logits.append(model(x[s], age[s]))
    logits = torch.cat(logits)
    crit = torch.nn.BCEWithLogitsLoss(pos_weight=torch.tensor(pos_weight))
    loss = crit(logits, y.reshape(-1))
    loss.backward()
    g = {n: p.grad.detach().clone() for n, p in model.named_parameters()
         if p.grad is not None}
    o = pack_offsets(model)
    packed = torch.zeros(o["npack"])

    def put(name, t):
        a, b = o[name]
        packed[a:b] = t.reshape(-1)

    put("w1", g["conv1.weight"]); put("b1", g["conv1.bias"])
    put("w2", g["conv2.weight"]); put("b2", g["conv2.bias"])
    put("wih1", g["lstm.weight_ih_l0"]); put("whh1", g["lstm.weight_hh_l0"])
    put("bl1", g["lstm.bias_ih_l0"])  # == bias_hh grad
    put("wih2", g["lstm.weight_ih_l1"]); put("whh2", g["lstm.weight_hh_l1"])
    put("bl2", g["lstm.bias_ih_l1"])
    put("outw", g["out.weight"]); put("outb", g["out.bias"])
    assert torch.allclose(g["lstm.bias_ih_l0"], g["lstm.bias_hh_l0"])
    return float(loss.item()), packed


@pytest.mark.gpu
class TestHipTraining:
    @pytest.mark.parametrize("variant,pos_weight", [("MyCNN5", 1.0),
                                                    ("MyCNN5", 3.5),
                                                    ("MyCNN2", 1.0),
                                                    ("MyCNN4", 2.0)])
    def test_grad_parity(self, variant, pos_weight):
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        torch.manual_seed(7)
        model = build_model(variant).eval()
        S, B = 3, 24
        x = torch.randn(S, B, model.IN_CHANNELS, 120)
        age = torch.full((S, B), 55.0)
        y = (torch.rand(S, B) < 0.3).float()
        ref_loss, ref_g = _torch_loss_and_grads(model, x, age, y, pos_weight)

        tr = MyCNNHipTrainer(model, device="cuda", pos_weight=pos_weight,
                             train_dropout=False)
        loss = tr.forward_backward(x.cuda(), age.cuda(), y.cuda())
        torch.cuda.synchronize()
        assert abs(loss - ref_loss) / max(abs(ref_loss), 1e-6) < 1e-3
        got = tr.grads.cpu()
        # compare per-slice for diagnosable failures
        o = tr.offsets
        for name in ("outw", "outb", "wih2", "whh2", "bl2", "wih1", "whh1",
                     "bl1", "w2", "b2", "w1", "b1"):
            a, b = o[name]
            torch.testing.assert_close(
                got[a:b], ref_g[a:b], rtol=2e-3, atol=2e-4,
                msg=lambda m, n=name: f"slice {n}: {m}")

    def test_grad_accumulation_is_additive(self):
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        torch.manual_seed(8)
        model = build_model("MyCNN5").eval()
        tr = MyCNNHipTrainer(model, device="cuda", train_dropout=False)
        x = torch.randn(2, 16, 10, 120).cuda()
        age = torch.full((2, 16), 65.0).cuda()
        y = (torch.rand(2, 16) < 0.5).float().cuda()
        tr.forward_backward(x, age, y)
        torch.cuda.synchronize()
        g1 = tr.grads.clone()
        tr.forward_backward(x, age, y)
        torch.cuda.synchronize()
        torch.testing.assert_close(tr.grads, 2 * g1, rtol=1e-4, atol=1e-6)

    def test_adam_matches_torch(self):
        from tskd_amd.train.hip_trainer import _load_train_lib, _p, _sp
        import ctypes
        lib = _load_train_lib()
        torch.manual_seed(9)
        n = 5242
        p0 = torch.randn(n)
        g0 = torch.randn(n)
        # torch oracle
        p_t = p0.clone().requires_grad_(True)
        opt = torch.optim.Adam([p_t], lr=1e-3)
        for step in range(3):
            p_t.grad = g0 * (step + 1)
            opt.step()
        # hip
        p_h = p0.clone().cuda()
        m = torch.zeros(n).cuda()
        v = torch.zeros(n).cuda()
        for step in range(3):
            g = (g0 * (step + 1)).cuda()
            rc = lib.tskd_train_adam(_p(p_h), _p(g), _p(m), _p(v), n,
                                     ctypes.c_float(1e-3),
                                     ctypes.c_float(0.9),
                                     ctypes.c_float(0.999),
                                     ctypes.c_float(1e-8), step + 1, _sp())
            assert rc == 0
            torch.cuda.synchronize()
            assert (g == 0).all()  # grads zeroed by the fused kernel
        torch.testing.assert_close(p_h.cpu(), p_t.detach(), rtol=1e-5,
                                   atol=1e-6)

    def test_training_reduces_loss(self):
        from tskd_amd.train.data import make_synthetic_labeled_windows
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        torch.manual_seed(10)
        model = build_model("MyCNN5").eval()
        tr = MyCNNHipTrainer(model, device="cuda", lr=3e-3)
        x, age, y = make_synthetic_labeled_windows(512, seed=4)
        x = torch.from_numpy(x).reshape(8, 64, 10, 120).cuda()
        age = torch.from_numpy(age).reshape(8, 64).cuda()
        y = torch.from_numpy(y).reshape(8, 64).cuda()
        losses = [tr.step(x, age, y) for _ in range(15)]
        assert losses[-1] < losses[0] * 0.9, losses

    def test_export_model_roundtrip(self):
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        torch.manual_seed(11)
        model = build_model("MyCNN5").eval()
        tr = MyCNNHipTrainer(model, device="cuda", lr=1e-2,
                             train_dropout=False)
        x = torch.randn(2, 32, 10, 120).cuda()
        age = torch.full((2, 32), 65.0).cuda()
        y = (torch.rand(2, 32) < 0.5).float().cuda()
        tr.step(x, age, y)
        m2 = tr.export_model()
        # exported torch model reproduces the packed forward
        wpack2 = pack_weights(m2)
        torch.testing.assert_close(wpack2, tr.wpack.cpu(), rtol=1e-6,
                                   atol=1e-7)


@pytest.mark.gpu
class TestDropout:
    def test_mask_exact_grad_parity(self):
        """Train-mode dropout (K5): extract the kernel's Bernoulli masks from
        the stash and replay them through a torch functional forward — loss
        and gradients must match exactly."""
        import torch.nn.functional as F

        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        torch.manual_seed(12)
        model = build_model("MyCNN5").eval()
        S, B = 2, 16
        x = torch.randn(S, B, 10, 120)
        age = torch.full((S, B), 65.0)
        y = (torch.rand(S, B) < 0.4).float()
        tr = MyCNNHipTrainer(model, device="cuda", pos_weight=1.0,
                             train_dropout=True, seed=99)
        loss = tr.forward_backward(x.cuda(), age.cuda(), y.cuda())
        torch.cuda.synchronize()
        stash = tr._last_stash_conv.cpu()  # (S*B, SC_SIZE)
        # stash offsets for MyCNN5: C1=111, C2=51, P1=55, LIN=25
        off_m1 = 4 * 111 + 51 + 4 * 55 + 25
        off_m2 = off_m1 + 4 * 55
        m1 = stash[:, off_m1:off_m1 + 4 * 55].reshape(S, B, 4, 55)
        m2 = stash[:, off_m2:off_m2 + 25].reshape(S, B, 25)
        drop_rate = float((m1 == 0).float().mean())
        assert 0.05 < drop_rate < 0.16  # p=0.1 Bernoulli

        # torch oracle with the SAME masks
        model.zero_grad()
        logits = []
        for s_ in range(S):
            h = torch.tanh(model.conv1(x[s_]))
            h = model.pool(h) * m1[s_]
            h = torch.tanh(model.conv2(h))
            h = model.pool(h).squeeze(1) * m2[s_]
            out, _ = model.lstm(h)
            z = model.out(out)
            scale = torch.relu(age[s_].unsqueeze(1) * 1e-8 + 1)
            logits.append((z * scale).squeeze(1))
        zz = torch.cat(logits)
        ref_loss = F.binary_cross_entropy_with_logits(zz, y.reshape(-1))
        ref_loss.backward()
        assert abs(loss - float(ref_loss)) / max(float(ref_loss), 1e-6) < 1e-3
        got = tr.grads.cpu()
        o = tr.offsets
        g = {n: p_.grad for n, p_ in model.named_parameters()
             if p_.grad is not None}
        for name, t in [("w1", g["conv1.weight"]), ("b1", g["conv1.bias"]),
                        ("w2", g["conv2.weight"]),
                        ("wih1", g["lstm.weight_ih_l0"]),
                        ("outw", g["out.weight"])]:
            a, b = o[name]
            torch.testing.assert_close(got[a:b], t.reshape(-1), rtol=2e-3,
                                       atol=2e-4,
                                       msg=lambda m, n=name: f"{n}: {m}")

    def test_masks_change_per_step(self):
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        model = build_model("MyCNN5").eval()
        tr = MyCNNHipTrainer(model, device="cuda", train_dropout=True)
        x = torch.randn(1, 8, 10, 120).cuda()
        age = torch.full((1, 8), 65.0).cuda()
        y = torch.zeros(1, 8).cuda()
        tr.forward_backward(x, age, y)
        torch.cuda.synchronize()
        m_a = tr._last_stash_conv[:, 740:].cpu().clone()
        tr.forward_backward(x, age, y)
        torch.cuda.synchronize()
        m_b = tr._last_stash_conv[:, 740:].cpu()
        assert not torch.equal(m_a, m_b)


@pytest.mark.gpu
class TestTrainingQuality:
    def test_hip_trained_model_beats_chance(self):
        """End-to-end quality: train on synthetic labeled windows with the
        HIP trainer, export to a torch module, verify held-out ROC-AUC."""
        from tskd_amd.train.data import make_synthetic_labeled_windows
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        from tskd_amd.train.report import classification_metrics, score_model
        torch.manual_seed(21)
        x, age, y = make_synthetic_labeled_windows(2048, pos_frac=0.3, seed=8)
        xv, av, yv = make_synthetic_labeled_windows(512, pos_frac=0.3, seed=9)
        model = build_model("MyCNN5").eval()
        n_pos = max((y == 1).sum(), 1)
        tr = MyCNNHipTrainer(model, device="cuda", lr=2e-3,
                             pos_weight=float((y == 0).sum() / n_pos))
        xb = torch.from_numpy(x).reshape(16, 128, 10, 120).cuda()
        ab = torch.from_numpy(age).reshape(16, 128).cuda()
        yb = torch.from_numpy(y).reshape(16, 128).cuda()
        for _ in range(30):
            tr.step(xb, ab, yb)
        trained = tr.export_model()
        probs = score_model(trained.cpu(), xv, av)
        rep = classification_metrics(yv, probs)
        assert rep["roc_auc"] > 0.8, rep


@pytest.mark.gpu
class TestBatchAccuracyKernel:
    """K14 fused metric kernel vs the torch oracle (utils.py:122-134)."""

    def test_matches_torch_oracle(self):
        from tskd_amd.train.metrics import compute_batch_accuracy
        torch.manual_seed(0)
        for n in (1, 63, 64, 257, 100_000):
            logits = torch.randn(n, device="cuda") * 3
            target = (torch.rand(n, device="cuda") < 0.4).float()
            got = compute_batch_accuracy(logits, target)
            pred = torch.sigmoid(logits).round()
            want = pred.eq(target).sum().float() * 100.0 / n
            assert abs(float(got) - float(want)) < 1e-3, (n, got, want)

    def test_zero_logit_rounds_down_and_nan_incorrect(self):
        from tskd_amd.train.metrics import compute_batch_accuracy
        # torch: round(sigmoid(0)) = round(0.5) = 0 (half-to-even);
        # NaN logits never equal the target
        logits = torch.tensor([0.0, 0.0, float("nan"), 5.0], device="cuda")
        target = torch.tensor([0.0, 1.0, 0.0, 1.0], device="cuda")
        got = float(compute_batch_accuracy(logits, target))
        pred = torch.sigmoid(logits).round()
        want = float(pred.eq(target).sum().float() * 100.0 / 4)
        assert abs(got - want) < 1e-6 and abs(got - 50.0) < 1e-6
