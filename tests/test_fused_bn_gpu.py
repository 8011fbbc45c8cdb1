"""GPU tests: fused Conv+BN+activation path vs the fp32 ATen composite."""
import numpy as np
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu

CL = torch.channels_last


@pytest.mark.parametrize("N,C,H,W,K,ks,st,pad,act", [
    (8, 64, 32, 32, 128, 3, 1, 1, "leaky"),
    (8, 64, 32, 32, 128, 4, 2, 1, "leaky"),
    (4, 32, 16, 16, 64, 3, 1, 1, "tanh"),
])
def test_fused_conv_bn_act_matches(N, C, H, W, K, ks, st, pad, act):
    from p2pvg_amd.ops.fused_norm import FusedSequential

    torch.manual_seed(0)
    act_mod = {"leaky": nn.LeakyReLU(0.2, inplace=True), "tanh": nn.Tanh()}[act]
    ref = nn.Sequential(
        nn.Conv2d(C, K, ks, st, pad), nn.BatchNorm2d(K), act_mod
    ).cuda()
    from p2pvg_amd.ops.conv import Conv2d
    from p2pvg_amd.ops.norm import BatchNorm2d

    act_mod2 = {"leaky": nn.LeakyReLU(0.2, inplace=True), "tanh": nn.Tanh()}[act]
    fused = FusedSequential(Conv2d(C, K, ks, st, pad), BatchNorm2d(K), act_mod2).cuda()
    fused.load_state_dict(ref.state_dict())

    x = torch.randn(N, C, H, W, device="cuda")
    xr = x.clone().requires_grad_()
    y_ref = ref(xr)
    g = torch.randn_like(y_ref)
    y_ref.backward(g)

    xh = x.bfloat16().contiguous(memory_format=CL).requires_grad_()
    y_hip = fused(xh)
    y_hip.backward(g.bfloat16().contiguous(memory_format=CL))

    tol = 0.08
    err = (y_hip.float() - y_ref).abs().max().item()
    scale = y_ref.abs().max().item()
    assert err < tol * scale + tol, f"fwd err {err}"

    # running stats must match (the BN semantics contract)
    for a, b in (
        (ref[1].running_mean, fused[1].running_mean),
        (ref[1].running_var, fused[1].running_var),
    ):
        assert torch.allclose(a, b, rtol=0.05, atol=0.02), (a - b).abs().max()

    for name, a, b in [
        ("dx", xr.grad, xh.grad.float()),
        ("dw", ref[0].weight.grad, fused[0].weight.grad.float()),
        ("dgamma", ref[1].weight.grad, fused[1].weight.grad.float()),
        ("dbeta", ref[1].bias.grad, fused[1].bias.grad.float()),
    ]:
        s = a.abs().max().item() + 1e-6
        e = (a - b).abs().max().item()
        assert e < tol * s + tol, f"{name}: err {e} scale {s}"


def test_fused_eval_mode_uses_running_stats():
    from p2pvg_amd.ops.conv import Conv2d
    from p2pvg_amd.ops.fused_norm import FusedSequential
    from p2pvg_amd.ops.norm import BatchNorm2d

    torch.manual_seed(1)
    m = FusedSequential(Conv2d(16, 32, 3, 1, 1), BatchNorm2d(32),
                        nn.LeakyReLU(0.2)).cuda()
    x = torch.randn(4, 16, 16, 16, device="cuda")
    # a few training steps to move running stats
    for _ in range(3):
        m(x.bfloat16().contiguous(memory_format=CL))
    m.eval()
    with torch.no_grad():
        y = m(x.bfloat16().contiguous(memory_format=CL))
        ref_m = nn.Sequential(nn.Conv2d(16, 32, 3, 1, 1), nn.BatchNorm2d(32),
                              nn.LeakyReLU(0.2)).cuda()
        ref_m.load_state_dict(m.state_dict())
        ref_m.eval()
        y_ref = ref_m(x)
    err = (y.float() - y_ref).abs().max().item()
    assert err < 0.08 * y_ref.abs().max().item() + 0.08


def test_model_step_fused_path_learns():
    from p2pvg_amd.core import Config
    from p2pvg_amd.models import P2PModel

    cfg = Config(dataset="bair", backbone="dcgan", channels=3, batch_size=4,
                 max_seq_len=5, g_dim=64, z_dim=8, rnn_size=128, device="cuda",
                 skip_prob=0.0, dtype="bf16")
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg).to("cuda").to(memory_format=torch.channels_last)
    x = torch.rand(5, 4, 3, 64, 64, device="cuda")
    first = last = None
    for i in range(6):
        model.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            losses = model(x, 0, 4)
        if first is None:
            first = float(losses[0])
        last = float(losses[0])
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses)
    assert last < first, f"{first} -> {last}"
