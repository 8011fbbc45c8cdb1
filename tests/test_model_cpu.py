"""Model-level CPU tests: training step semantics, losses, generation."""
import numpy as np
import pytest
import torch

from p2pvg_amd.core import Config
from p2pvg_amd.models import P2PModel


def make_batch(cfg, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.rand(
        cfg.max_seq_len, cfg.batch_size, cfg.channels, cfg.image_width, cfg.image_width,
        generator=g,
    )


def test_forward_returns_four_losses(tiny_cfg):
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(tiny_cfg)
    x = make_batch(tiny_cfg)
    mse, kld, cpc, align = model(x, 0, len(x) - 1)
    for v in (mse, kld, cpc, align):
        assert torch.is_tensor(v) and v.dim() == 0
        assert torch.isfinite(v)
    assert mse > 0
    assert kld > 0
    assert cpc > 0  # cp step always processed


def test_training_decreases_loss(tiny_cfg):
    torch.manual_seed(0)
    np.random.seed(0)
    tiny_cfg.skip_prob = 0.0
    tiny_cfg.lr = 1e-3
    model = P2PModel(tiny_cfg)
    x = make_batch(tiny_cfg)
    first = None
    last = None
    for i in range(8):
        model.zero_grad(set_to_none=False)
        mse, _, _, _ = model(x, 0, len(x) - 1)
        if first is None:
            first = mse.item()
        last = mse.item()
    assert last < first, f"mse did not decrease: {first} -> {last}"


def test_skip_gate_semantics(tiny_cfg):
    """Skip gate must never skip i==1 or the cp step, and bounds skips."""
    torch.manual_seed(0)
    tiny_cfg.skip_prob = 1.0  # always want to skip where legal
    model = P2PModel(tiny_cfg)
    x = make_batch(tiny_cfg)
    np.random.seed(0)
    mse, kld, cpc, align = model(x, 0, len(x) - 1)
    # cpc computed means cp step ran
    assert cpc > 0


def test_generate_modes_and_lengths(tiny_cfg):
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(tiny_cfg)
    model.eval()
    x = make_batch(tiny_cfg)
    for mode in ("full", "posterior", "prior"):
        for length in (5, 8, 12):
            out = model.p2p_generate(x, length, length - 1, model_mode=mode)
            assert len(out) == length
            for f in out:
                assert f.shape == x[0].shape


def test_generate_skip_frame_inserts_zeros(tiny_cfg):
    torch.manual_seed(0)
    model = P2PModel(tiny_cfg)
    model.eval()
    tiny_cfg.skip_prob = 1.0
    x = make_batch(tiny_cfg)
    np.random.seed(3)
    out = model.p2p_generate(x, 12, 11, model_mode="full", skip_frame=True)
    assert len(out) == 12
    zeros = sum(1 for f in out if f.abs().sum() == 0)
    assert zeros >= 1  # some frames skipped as placeholders


def test_mlp_backbone_h36m_path():
    cfg = Config(
        dataset="h36m", backbone="mlp", batch_size=2, max_seq_len=8, delta_len=1,
        g_dim=32, z_dim=4, rnn_size=32, device="cpu",
    )
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg)
    pose_3d = torch.randn(8, 2, 17, 3)
    pose_2d = pose_3d[..., :2].clone()
    losses = model((pose_2d, pose_3d, [0, 1]), 0, 7)
    assert all(torch.isfinite(v) for v in losses)


@pytest.mark.parametrize("backbone,width", [("dcgan", 64), ("dcgan", 128), ("vgg", 64), ("vgg", 128)])
def test_all_backbones_shapes(backbone, width):
    from p2pvg_amd.models.backbones import get_backbone

    enc_cls, dec_cls = get_backbone(backbone, width, "bair")
    enc, dec = enc_cls(32, 3), dec_cls(32, 3)
    x = torch.randn(2, 3, width, width)
    latent, skips = enc(x)
    assert latent.shape == (2, 32)
    out = dec([latent, skips])
    assert out.shape == x.shape


def test_multi_cp_and_loop_generation(tiny_cfg):
    """Chained segment generation through multiple control points and the
    A->B->A loop (reference README showcase capabilities)."""
    import generate as gen_cli

    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(tiny_cfg)
    model.eval()
    x = make_batch(tiny_cfg)
    frames = [x[0], x[3], x[7]]
    out = gen_cli.multi_cp_generate(model, frames, seg_len=5)
    # two segments of 5 sharing the joint frame
    assert len(out) == 9
    loop = gen_cli.loop_generate(model, x[0], x[7], seg_len=4)
    assert len(loop) == 7
    assert torch.equal(loop[0], x[0])


def test_conv_weight_shadows_track_weight_version():
    """The per-step bf16 weight shadows must refill when the weight changes
    in place (version counter) and hold the right flip/transpose forms."""
    import torch

    from p2pvg_amd.ops.conv import Conv2d, ConvTranspose2d, _conv_shadows

    torch.manual_seed(0)
    c = Conv2d(8, 16, 3, stride=1, padding=1, bias=False)
    sh = _conv_shadows(c)
    assert sh["f"].dtype == torch.bfloat16
    torch.testing.assert_close(sh["f"].float(), c.weight.detach().bfloat16().float())
    torch.testing.assert_close(
        sh["b"].float(),
        c.weight.detach().bfloat16().flip(2, 3).transpose(0, 1).float(),
    )

    with torch.no_grad():
        c.weight.add_(1.0)
    sh2 = _conv_shadows(c)
    assert sh2 is sh, "shadow storage must be reused"
    torch.testing.assert_close(sh["f"].float(), c.weight.detach().bfloat16().float())

    ct = ConvTranspose2d(8, 4, 4, stride=2, padding=1, bias=False)
    sht = _conv_shadows(ct)
    torch.testing.assert_close(
        sht["f"].float(), ct.weight.detach().bfloat16().transpose(0, 1).float()
    )
    torch.testing.assert_close(sht["b"].float(), ct.weight.detach().bfloat16().float())


def test_zero_grads_stable_buffers(tiny_cfg):
    """zero_grads materializes every .grad once and keeps buffer identity
    stable across steps (hipGraph capture + fused Adam rely on this)."""
    import torch

    from p2pvg_amd.models import P2PModel

    model = P2PModel(tiny_cfg)
    model.zero_grads()
    ids = {n: id(p.grad) for n, p in model.named_parameters()}
    assert all(p.grad is not None for p in model.parameters())
    x = torch.rand(4, tiny_cfg.batch_size, tiny_cfg.channels,
                   tiny_cfg.image_width, tiny_cfg.image_width)
    model(x, 0, 3)
    model.zero_grads()
    for n, p in model.named_parameters():
        assert id(p.grad) == ids[n], f"{n}: grad buffer identity changed"
        assert float(p.grad.abs().max()) == 0.0


def test_gauss_stacked_cache_refreshes_on_init_hidden():
    """The fused-head weight cache [Wm; Wl] must follow the params: stale
    until init_hidden marks it dirty (one refresh per step/sequence)."""
    import torch

    from p2pvg_amd.models.lstm import gaussian_lstm

    stack = gaussian_lstm(258, 10, 256, 1, 4)
    stack.init_hidden(4)
    ws, bs = stack._stacked()
    assert ws.shape == (20, 256) and bs.shape == (20,)
    assert torch.equal(ws[:10], stack.mu_net.weight)
    assert torch.equal(ws[10:], stack.logvar_net.weight)

    with torch.no_grad():
        stack.mu_net.weight.add_(1.0)
    # same step: cache intentionally NOT refreshed (weights only move at
    # optimizer.step(), which is always followed by a new init_hidden)
    ws2, _ = stack._stacked()
    assert ws2 is ws and not torch.equal(ws[:10], stack.mu_net.weight)

    stack.init_hidden(4)  # new step/sequence -> refresh
    ws3, bs3 = stack._stacked()
    assert ws3 is ws  # persistent buffer, stable storage for hipGraph replay
    assert torch.equal(ws[:10], stack.mu_net.weight)
    assert torch.equal(bs3[:10], stack.mu_net.bias)


def test_graph_warmup_snapshot_restore_cpu(tiny_cfg):
    """_snapshot_train_state/_restore_train_state (the zero-net-update
    capture-warmup mechanism) restore weights AND optimizer state bitwise,
    and zero Adam state created after the snapshot."""
    import torch

    from p2pvg_amd.models import P2PModel
    from p2pvg_amd.runtime import GraphedTrainStep

    torch.manual_seed(0)
    model = P2PModel(tiny_cfg)
    stepper = GraphedTrainStep(model)
    x = torch.rand(tiny_cfg.max_seq_len, tiny_cfg.batch_size,
                   tiny_cfg.channels, tiny_cfg.image_width,
                   tiny_cfg.image_width)

    def one_step():
        model.zero_grad(set_to_none=False)
        model(x, 0, len(x) - 1)

    one_step()  # creates Adam state
    snap = stepper._snapshot_train_state()
    ref_sd = {k: v.clone() for k, v in model.state_dict().items()}

    one_step()
    one_step()
    changed = any(not torch.equal(v, model.state_dict()[k])
                  for k, v in ref_sd.items() if v.is_floating_point())
    assert changed, "steps did not change weights; test is vacuous"

    stepper._restore_train_state(snap)
    for k, v in model.state_dict().items():
        assert torch.equal(v, ref_sd[k]), f"weight not restored: {k}"
    for name, opt in stepper._optimizers():
        msnap = snap[1][name]
        for p, st in opt.state.items():
            for kk, vv in st.items():
                if torch.is_tensor(vv):
                    assert torch.equal(vv, msnap[id(p)][kk]), \
                        f"opt state not restored: {name}/{kk}"

    # state created AFTER an early snapshot gets reset to zeros on restore
    model2 = P2PModel(tiny_cfg)
    stepper2 = GraphedTrainStep(model2)
    snap2 = stepper2._snapshot_train_state()  # before any optimizer state
    model2.zero_grad(set_to_none=False)
    model2(x, 0, len(x) - 1)
    stepper2._restore_train_state(snap2)
    for _, opt in stepper2._optimizers():
        for _, st in opt.state.items():
            for kk, vv in st.items():
                if torch.is_tensor(vv) and vv.is_floating_point():
                    assert vv.abs().sum() == 0, f"fresh state not zeroed: {kk}"


def test_fused_heads_dispatch_policy(monkeypatch):
    """_fused_heads gate: cuda-only, batch <= 512 (hipBLASLt wins above,
    profiles/MEASUREMENTS.md note 3), P2PVG_LSTM_HEADS=0 kill switch."""
    import torch

    from p2pvg_amd.models.lstm import gaussian_lstm

    stack = gaussian_lstm(258, 10, 256, 1, 4)
    cpu_small = torch.randn(4, 128)
    assert not stack._fused_heads(cpu_small)          # CPU: never
    assert not stack._fused_heads((cpu_small,) * 4)   # tuple form too

    class FakeCuda:
        is_cuda = True

        def __init__(self, rows):
            self.shape = (rows, 128)

    from p2pvg_amd import ops

    if ops.hip_available():  # CPU containers build the ext; gate if absent
        assert stack._fused_heads((FakeCuda(512),))
        assert not stack._fused_heads((FakeCuda(513),))   # large-batch: GEMM heads
        monkeypatch.setenv("P2PVG_LSTM_HEADS", "0")
        assert not stack._fused_heads((FakeCuda(4),))
