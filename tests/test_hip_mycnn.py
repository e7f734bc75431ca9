"""HIP kernel numerics: fused conv+LSTM inference vs the PyTorch fp32 oracle.

Each GPU test compares the hand-written CDNA4 kernels against the plain
PyTorch fp32 eager model (tskd_amd.models.mycnn, itself verified against an
independent numpy implementation in test_models.py).
"""

import pytest
import torch

from tskd_amd.models import build_model
from tskd_amd.ops import MyCNNEngine, hip_available


def _x(s, n, c, seed=0, dtype=torch.float32):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(s, n, c, 120, generator=g, dtype=torch.float32).to(dtype)


def _oracle(model, x, age=None, sigmoid=False):
    """CPU fp32 truth, sequence by sequence (batch-as-time per sequence)."""
    s, n = x.shape[0], x.shape[1]
    outs = []
    with torch.no_grad():
        for i in range(s):
            a = age[i].float() if age is not None else torch.zeros(n)
            y = model(x[i].float(), a)
            outs.append(torch.sigmoid(y) if sigmoid else y)
    return torch.stack(outs)


def test_build_cross_compiles():
    # hipcc works without a GPU: the .so must build in this container.
    assert hip_available()


@pytest.mark.gpu
class TestHipForward:
    @pytest.mark.parametrize("variant,n", [("MyCNN5", 64), ("MyCNN5", 37),
                                           ("MyCNN5", 1), ("MyCNN2", 64),
                                           ("MyCNN4", 64)])
    def test_fp32_parity(self, variant, n):
        # n covers CHUNK-multiple, odd, and single-step scans
        m = build_model(variant).eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(3, n, m.IN_CHANNELS, seed=1)
        ref = _oracle(m, x)
        got = eng.forward(x.cuda()).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-5)

    def test_mfma_fragment_map(self):
        """Ground-truth check of the v_mfma_f32_16x16x32_bf16 A/B/C lane maps
        (asymmetric operands so a transposed map cannot pass — §5.4 rule 16)."""
        import ctypes
        from tskd_amd.ops import _load_lib, _stream_ptr
        lib = _load_lib()
        g = torch.Generator().manual_seed(11)
        A = (torch.randn(16, 32, generator=g) * 0.5).to(torch.bfloat16)
        B = (torch.randn(32, 16, generator=g) * 0.5 +
             torch.arange(16).float() * 0.01).to(torch.bfloat16)
        Ad, Bd = A.cuda(), B.cuda()
        Cd = torch.empty(16, 16, dtype=torch.float32, device="cuda")
        rc = lib.tskd_debug_mfma16x16x32(
            ctypes.c_void_p(Ad.data_ptr()), ctypes.c_void_p(Bd.data_ptr()),
            ctypes.c_void_p(Cd.data_ptr()), _stream_ptr())
        assert rc == 0
        torch.cuda.synchronize()
        ref = A.float() @ B.float()
        torch.testing.assert_close(Cd.cpu(), ref, rtol=2e-2, atol=2e-2)

    def test_bf16_input_mfma_conv(self):
        # bf16 path runs conv1 on MFMA (im2col-GEMM); weights + inputs are
        # bf16-rounded, accumulation fp32.
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(2, 32, 10, seed=2, dtype=torch.bfloat16)
        # Oracle sees the same bf16-rounded windows (fp32 weights).
        ref = _oracle(m, x.float())
        got = eng.forward(x.cuda()).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-2, atol=2e-2)

    @pytest.mark.parametrize("variant", ["MyCNN5", "MyCNN2", "MyCNN4"])
    def test_bf16_conv_features_all_variants(self, variant):
        m = build_model(variant).eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 16, m.IN_CHANNELS, seed=9, dtype=torch.bfloat16)
        with torch.no_grad():
            f = m.pool(torch.tanh(m.conv2(m.pool(torch.tanh(m.conv1(x[0].float()))))))
            ref = f.view(16, -1)
        got = eng.conv_features(x.cuda()).reshape(16, -1).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-2, atol=2e-2)

    def test_long_sequence_accumulation(self):
        # The LSTM scan is sequential over N=1024: accumulated fp error must
        # stay tiny (state is bounded by tanh/sigmoid saturation).
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 1024, 10, seed=3)
        ref = _oracle(m, x)
        got = eng.forward(x.cuda()).cpu()
        torch.testing.assert_close(got, ref, rtol=5e-4, atol=5e-5)

    def test_sequences_independent(self):
        # (S, N) call == S separate (N,) calls: no cross-sequence state leak.
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(4, 32, 10, seed=4).cuda()
        batched = eng.forward(x)
        singles = torch.stack([eng.forward(x[i]) for i in range(4)])
        torch.testing.assert_close(batched, singles, rtol=1e-6, atol=1e-7)

    def test_age_gate_and_sigmoid(self):
        m = build_model("MyCNN4").eval()  # AGE_EPS=1e-4: age NOT inert
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 16, 10, seed=5)
        age = torch.full((1, 16), 70.0)
        ref = _oracle(m, x, age, sigmoid=True)
        got = eng.forward(x.cuda(), age.cuda(), apply_sigmoid=True).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-5)
        assert ((got >= 0) & (got <= 1)).all()

    def test_conv_features_parity(self):
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 8, 10, seed=6)
        with torch.no_grad():
            f = m.pool(torch.tanh(m.conv2(m.pool(torch.tanh(m.conv1(x[0]))))))
            ref = f.view(-1, 25)
        got = eng.conv_features(x.cuda()).reshape(-1, 25).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-5, atol=2e-6)

    def test_reference_checkpoint_on_gpu(self):
        import os
        path = "/root/repo/tests/data/ref_mycnn5.pth"
        if not os.path.exists(path):
            pytest.skip("golden checkpoint not present")
        from tskd_amd.models import load_checkpoint
        m = load_checkpoint(path)
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 16, 10, seed=7)
        ref = _oracle(m, x)
        got = eng.forward(x.cuda()).cpu()
        torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-5)


@pytest.mark.gpu
class TestGraphedForward:
    def test_graph_matches_eager(self):
        from tskd_amd.ops import GraphedForward
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        g = GraphedForward(eng, s=8, n=16, dtype=torch.bfloat16)
        x = _x(8, 16, 10, seed=21, dtype=torch.bfloat16).cuda()
        age = torch.full((8, 16), 65.0, device="cuda")
        g.x.copy_(x)
        g.age.copy_(age)
        out = g.replay().clone()
        torch.cuda.synchronize()
        ref = eng.forward(x, age, apply_sigmoid=True)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
        # replay with NEW data must track the new input
        x2 = _x(8, 16, 10, seed=22, dtype=torch.bfloat16).cuda()
        g.x.copy_(x2)
        out2 = g.replay().clone()
        torch.cuda.synchronize()
        ref2 = eng.forward(x2, age, apply_sigmoid=True)
        torch.testing.assert_close(out2, ref2, rtol=1e-5, atol=1e-6)
        assert not torch.allclose(out, out2)

    def test_gather_into_graph_buffer(self):
        from tskd_amd.engine import StreamEngine
        from tskd_amd.ops import GraphedForward
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        se = StreamEngine(4, 10, ring_grid=1024, fs=125.0, device="cuda")
        raw = torch.randn(4, 8, int(125 * 60 * 16), device="cuda")
        se.ingest_dense(raw, chan_map=list(range(8)))
        g = GraphedForward(eng, s=4, n=1, dtype=torch.bfloat16)
        w = se.windows(batch=1, stride=12, dtype=torch.bfloat16, out=g.x)
        assert w.data_ptr() == g.x.data_ptr()
        out = g.replay()
        torch.cuda.synchronize()
        ref = eng.forward(g.x, g.age, apply_sigmoid=True)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)


@pytest.mark.gpu
class TestTimelastLayout:
    def test_tlast_matches_std(self):
        from tskd_amd.ops import alloc_windows
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(2, 32, 10, seed=31, dtype=torch.bfloat16).cuda()
        ref = eng.forward(x)  # staged-layout path
        xt = alloc_windows(2, 32, 10, timelast=True, dtype=torch.bfloat16)
        xt.copy_(x.transpose(-1, -2))
        got = eng.forward(xt)  # LDS-free timelast path
        torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-5)

    def test_tlast_without_slack_is_copied_safely(self):
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        x = _x(1, 8, 10, seed=32, dtype=torch.bfloat16).cuda()
        xt = x.transpose(-1, -2).contiguous()  # no slack tag -> wrapper copies
        ref = eng.forward(x)
        got = eng.forward(xt)
        torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-5)

    def test_gather_timelast(self):
        from tskd_amd.engine import StreamEngine
        se_c = StreamEngine(3, 10, ring_grid=1024, fs=25.0, device="cpu")
        se_g = StreamEngine(3, 10, ring_grid=1024, fs=25.0, device="cuda")
        raw = torch.randn(3, 8, int(25 * 60 * 16))
        se_c.ingest_dense(raw, chan_map=list(range(8)))
        se_g.ingest_dense(raw.cuda(), chan_map=list(range(8)))
        wc = se_c.windows(batch=2, stride=12, timelast=True)
        wg = se_g.windows(batch=2, stride=12, timelast=True)
        torch.cuda.synchronize()
        assert wc.shape == (3, 2, 120, 10)
        torch.testing.assert_close(wg.cpu(), wc, rtol=1e-4, atol=1e-5)

    def test_graphed_timelast_pipeline(self):
        from tskd_amd.engine import StreamEngine
        from tskd_amd.ops import GraphedForward
        m = build_model("MyCNN5").eval()
        eng = MyCNNEngine(m, device="cuda")
        se = StreamEngine(4, 10, ring_grid=1024, fs=125.0, device="cuda")
        se.ingest_dense(torch.randn(4, 8, int(125 * 60 * 16), device="cuda"),
                        chan_map=list(range(8)))
        g = GraphedForward(eng, s=4, n=1, dtype=torch.bfloat16, timelast=True)
        se.windows(batch=1, stride=12, dtype=torch.bfloat16, out=g.x,
                   timelast=True)
        out = g.replay().clone()
        torch.cuda.synchronize()
        w_std = se.windows(batch=1, stride=12, dtype=torch.bfloat16)
        ref = eng.forward(w_std, g.age, apply_sigmoid=True)
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
