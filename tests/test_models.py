"""Model-core tests: shapes, the LSTM-over-batch quirk, variant geometry."""

import numpy as np
import pytest
import torch

from tskd_amd.models import MyCNN2, MyCNN3, MyCNN4, MyCNN5, build_model


def _win(n, c, l, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, c, l, generator=g)


class TestShapes:
    def test_mycnn5_shapes(self):
        m = MyCNN5().eval()
        x = _win(8, 10, 120)
        age = torch.full((8,), 65.0)
        y = m(x, age)
        assert y.shape == (8,)

    def test_mycnn5_intermediate_geometry(self):
        # (N,10,120) -> conv1 k10 -> 111 -> pool(3,2) -> 55 -> conv2 k5 -> 51
        # -> pool -> 25 (SURVEY.md §2.3 forward semantics)
        m = MyCNN5().eval()
        x = _win(2, 10, 120)
        a = torch.tanh(m.conv1(x))
        assert a.shape == (2, 4, 111)
        a = m.pool(a)
        assert a.shape == (2, 4, 55)
        a = torch.tanh(m.conv2(a))
        assert a.shape == (2, 1, 51)
        a = m.pool(a)
        assert a.shape == (2, 1, 25)

    @pytest.mark.parametrize("cls,c", [(MyCNN2, 7), (MyCNN3, 7), (MyCNN4, 10)])
    def test_small_variants(self, cls, c):
        m = cls().eval()
        y = m(_win(4, c, 120), torch.full((4,), 50.0))
        assert y.shape == (4,)

    def test_build_model(self):
        assert isinstance(build_model("MyCNN2"), MyCNN2)
        with pytest.raises(ValueError):
            build_model("nope")


class TestLstmOverBatchQuirk:
    """The 2-D LSTM input makes the batch axis the time axis: window i's
    output depends on windows 0..i-1 (reference bin/models.py:30)."""

    def test_batch_order_matters(self):
        m = MyCNN5().eval()
        x = _win(16, 10, 120)
        age = torch.full((16,), 65.0)
        with torch.no_grad():
            y = m(x, age)
            perm = torch.randperm(16, generator=torch.Generator().manual_seed(1))
            y_perm = m(x[perm], age)
        # Same multiset of windows, different order => different outputs.
        assert not torch.allclose(torch.sort(y).values, torch.sort(y_perm).values)

    def test_prefix_stability(self):
        # Window i's output is identical whether or not later windows exist.
        m = MyCNN5().eval()
        x = _win(16, 10, 120)
        age = torch.full((16,), 65.0)
        with torch.no_grad():
            y_full = m(x, age)
            y_half = m(x[:8], age[:8])
        assert torch.allclose(y_full[:8], y_half, atol=1e-6)

    def test_matches_explicit_unbatched_lstm(self):
        m = MyCNN5().eval()
        x = _win(6, 10, 120)
        with torch.no_grad():
            feat = m.pool(torch.tanh(m.conv2(m.pool(torch.tanh(m.conv1(x))))))
            feat = feat.view(-1, 25)
            # Explicit (seq, batch=1, feat) call must equal the 2-D call.
            out3d, _ = m.lstm(feat.unsqueeze(1))
            out2d, _ = m.lstm(feat)
        assert torch.allclose(out3d.squeeze(1), out2d, atol=1e-7)


class TestAgeGate:
    def test_age_inert_in_mycnn5(self):
        # relu(age*1e-8 + 1) ~= 1.0 for clinical ages => age is inert.
        m = MyCNN5().eval()
        x = _win(4, 10, 120)
        with torch.no_grad():
            y0 = m(x, torch.full((4,), 20.0))
            y1 = m(x, torch.full((4,), 80.0))
        assert torch.allclose(y0, y1, atol=1e-5)

    def test_eval_dropout_identity(self):
        m = MyCNN5().eval()
        x = _win(4, 10, 120)
        a = torch.full((4,), 65.0)
        with torch.no_grad():
            assert torch.equal(m(x, a), m(x, a))


class TestNumpyParity:
    """Independent numpy re-implementation of the MyCNN5 forward — guards the
    torch-truth itself so HIP-kernel tests inherit a verified oracle."""

    def test_numpy_forward_matches(self):
        m = MyCNN5().double().eval()
        n = 5
        x = _win(n, 10, 120).double()
        age = torch.full((n,), 65.0).double()

        sd = {k: v.detach().numpy() for k, v in m.state_dict().items()}

        def conv1d(x, w, b):
            co, ci, k = w.shape
            nn_, _, l = x.shape
            out = np.zeros((nn_, co, l - k + 1))
            for t in range(l - k + 1):
                out[:, :, t] = np.einsum("nik,oik->no", x[:, :, t:t + k], w) + b
            return out

        def maxpool(x, k=3, s=2):
            l = x.shape[-1]
            nsteps = (l - k) // s + 1
            return np.stack([x[..., i * s:i * s + k].max(-1) for i in range(nsteps)], -1)

        h = np.tanh(conv1d(x.numpy(), sd["conv1.weight"], sd["conv1.bias"]))
        h = maxpool(h)
        h = np.tanh(conv1d(h, sd["conv2.weight"], sd["conv2.bias"]))
        h = maxpool(h).reshape(n, 25)

        # 2-layer LSTM over the batch axis (gate order i,f,g,o).
        def sigmoid(z):
            return 1.0 / (1.0 + np.exp(-z))

        inp = h
        for layer in range(2):
            wi = sd[f"lstm.weight_ih_l{layer}"]
            wh = sd[f"lstm.weight_hh_l{layer}"]
            bi = sd[f"lstm.bias_ih_l{layer}"]
            bh = sd[f"lstm.bias_hh_l{layer}"]
            hs, cs = np.zeros(16), np.zeros(16)
            outs = []
            for t in range(n):
                gates = wi @ inp[t] + bi + wh @ hs + bh
                i_, f_, g_, o_ = np.split(gates, 4)
                cs = sigmoid(f_) * cs + sigmoid(i_) * np.tanh(g_)
                hs = sigmoid(o_) * np.tanh(cs)
                outs.append(hs.copy())
            inp = np.stack(outs)

        logits = inp @ sd["out.weight"].T + sd["out.bias"]
        scale = np.maximum(age.numpy()[:, None] * 1e-8 + 1, 0.0)
        ref = (logits * scale).squeeze(1)

        with torch.no_grad():
            y = m(x, age).numpy()
        np.testing.assert_allclose(y, ref, rtol=1e-10, atol=1e-12)
