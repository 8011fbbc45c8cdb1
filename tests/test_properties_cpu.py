"""Property-based tests (hypothesis) for the host-side invariants the GPU
path depends on: the skip-gate step plan (reference models/p2p_model.py:209-222
semantics), config round-trips, and metric identities."""
import numpy as np
import pytest
import torch

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402


def _model(skip_prob, n_past, max_seq_len):
    from p2pvg_amd.core import Config
    from p2pvg_amd.models import P2PModel

    cfg = Config(dataset="mnist", backbone="dcgan", channels=1, batch_size=2,
                 max_seq_len=max_seq_len, g_dim=16, z_dim=4, rnn_size=32,
                 skip_prob=skip_prob, n_past=n_past, device="cpu")
    return P2PModel(cfg)


@settings(max_examples=40, deadline=None)
@given(
    seed=st.integers(0, 2**31 - 1),
    skip_prob=st.floats(0.0, 0.9),
    n_past=st.integers(1, 3),
    seq_len=st.integers(4, 30),
)
def test_plan_step_invariants(seed, skip_prob, n_past, seq_len):
    model = _model(skip_prob, n_past, max(seq_len, 4))
    np.random.seed(seed)
    plan = model.plan_step(seq_len)
    cp = seq_len - 1

    # processed steps: strictly increasing, within [1, cp]
    assert list(plan.proc) == sorted(set(plan.proc))
    assert all(1 <= i <= cp for i in plan.proc)
    # never skips step 1 or the control-point step
    assert 1 in plan.proc
    assert cp in plan.proc or cp == 1
    # skip budget (reference gate: skip only while skip_count < T*skip_prob,
    # so the count may reach ceil(T*skip_prob))
    skips = cp - len(plan.proc)
    assert skips == 0 or (skips - 1) < seq_len * skip_prob
    # time signals consistent with the processed indices
    assert len(plan.tun) == len(plan.proc) == len(plan.dts)
    prev = 0
    for i, t, d in zip(plan.proc, plan.tun, plan.dts):
        assert abs(t - (cp - i + 1) / cp) < 1e-5
        assert abs(d - (i - prev) / cp) < 1e-5
        prev = i
    # delta_time sums to cp/cp = 1 exactly when the walk ends at cp
    if plan.proc[-1] == cp:
        assert abs(float(sum(plan.dts)) - 1.0) < 1e-5
    # graph key is hashable + stable
    assert plan.graph_key == (seq_len, len(plan.proc), plan.unpack)
    hash(plan.graph_key)


@settings(max_examples=25, deadline=None)
@given(
    skip_prob=st.floats(0.0, 0.9),
    beta=st.floats(1e-6, 1.0),
    batch_size=st.integers(1, 512),
    g_dim=st.sampled_from([64, 128]),
)
def test_config_roundtrip(skip_prob, beta, batch_size, g_dim):
    from p2pvg_amd.core import Config

    cfg = Config(skip_prob=skip_prob, beta=beta, batch_size=batch_size,
                 g_dim=g_dim)
    cfg2 = Config.from_dict(cfg.to_dict())
    assert cfg2.to_dict() == cfg.to_dict()


@settings(max_examples=10, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_ssim_identity_and_range(seed):
    from p2pvg_amd.utils import end_frame_ssim
    from p2pvg_amd.utils.metrics import ssim

    g = torch.Generator().manual_seed(seed)
    x = torch.rand(2, 3, 32, 32, generator=g)
    y = torch.rand(2, 3, 32, 32, generator=g)
    s_same = ssim(x, x)
    assert torch.all(s_same > 0.999)
    s_diff = ssim(x, y)
    assert torch.all(s_diff <= 1.0 + 1e-6) and torch.all(s_diff >= -1.0 - 1e-6)
    assert torch.all(s_diff < s_same)
    e = end_frame_ssim([x, y], y)
    assert torch.all(e > 0.999)
