"""GPU tests: full autograd (dgrad/wgrad/bias) of the gfx950 conv kernels vs
the fp32 ATen reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def _run_pair(fn_ref, fn_hip, x, w, b, tol):
    x_ref = x.clone().requires_grad_()
    w_ref = w.clone().requires_grad_()
    b_ref = b.clone().requires_grad_() if b is not None else None
    out_ref = fn_ref(x_ref, w_ref, b_ref)
    g = torch.randn_like(out_ref)
    out_ref.backward(g)

    xl = x.bfloat16().contiguous(memory_format=CL).requires_grad_()
    wl = w.bfloat16().contiguous(memory_format=CL).requires_grad_()
    bl = b.clone().requires_grad_() if b is not None else None
    out_hip = fn_hip(xl, wl, bl)
    out_hip.backward(g.bfloat16().contiguous(memory_format=CL))

    scale = out_ref.abs().max().item() + 1e-6
    err = (out_hip.float() - out_ref).abs().max().item()
    assert err < tol * scale + tol, f"fwd err {err} scale {scale}"

    for name, a, bb in [
        ("dx", x_ref.grad, xl.grad.float()),
        ("dw", w_ref.grad, wl.grad.float()),
    ] + ([("db", b_ref.grad, bl.grad.float())] if b is not None else []):
        s = a.abs().max().item() + 1e-6
        e = (a - bb).abs().max().item()
        assert e < tol * s + tol, f"{name} err {e} scale {s}"


CONV_SHAPES = [
    (4, 64, 32, 32, 64, 3, 1, 1),
    (4, 128, 16, 16, 256, 3, 1, 1),
    (2, 3, 64, 64, 64, 3, 1, 1),
    (4, 64, 32, 32, 128, 4, 2, 1),
    (4, 512, 4, 4, 128, 4, 1, 0),
    (3, 80, 16, 16, 96, 3, 1, 1),
]


@pytest.mark.parametrize("N,C,H,W,K,ks,st,pad", CONV_SHAPES)
def test_conv2d_autograd_matches(N, C, H, W, K, ks, st, pad):
    from p2pvg_amd.ops.conv import Conv2dNHWCFn

    torch.manual_seed(0)
    x = torch.randn(N, C, H, W, device="cuda") * 0.5
    w = torch.randn(K, C, ks, ks, device="cuda") * (1.0 / (ks * ks * C) ** 0.5)
    b = torch.randn(K, device="cuda") * 0.1

    _run_pair(
        lambda xx, ww, bb: torch.nn.functional.conv2d(xx, ww, bb, stride=st, padding=pad),
        lambda xx, ww, bb: Conv2dNHWCFn.apply(xx, ww, bb, st, pad, 0, False, None, None, 0, 0)[0],
        x, w, b, tol=0.05,
    )


CONVT_SHAPES = [
    (4, 512, 8, 8, 256, 4, 2, 1),    # dcgan upconv
    (4, 128, 32, 32, 64, 4, 2, 1),
    (4, 128, 1, 1, 512, 4, 1, 0),    # decoder head 1->4
    (4, 64, 64, 64, 3, 3, 1, 1),     # vgg final convT
]


@pytest.mark.parametrize("N,Ci,H,W,Co,ks,st,pad", CONVT_SHAPES)
def test_convtranspose2d_autograd_matches(N, Ci, H, W, Co, ks, st, pad):
    from p2pvg_amd.ops.conv import ConvT2dNHWCFn

    torch.manual_seed(1)
    x = torch.randn(N, Ci, H, W, device="cuda") * 0.5
    w = torch.randn(Ci, Co, ks, ks, device="cuda") * (1.0 / (ks * ks * Ci) ** 0.5)
    b = torch.randn(Co, device="cuda") * 0.1

    _run_pair(
        lambda xx, ww, bb: torch.nn.functional.conv_transpose2d(
            xx, ww, bb, stride=st, padding=pad),
        lambda xx, ww, bb: ConvT2dNHWCFn.apply(xx, ww, bb, st, pad, 0, False, None, None, 0, 0)[0],
        x, w, b, tol=0.05,
    )


def test_module_dispatch_bf16():
    """The Conv2d module routes bf16 CUDA input to the HIP path and matches
    the fp32 reference within bf16 tolerance."""
    from p2pvg_amd.ops.conv import Conv2d

    torch.manual_seed(2)
    m = Conv2d(64, 128, 3, 1, 1).cuda()
    ref = torch.nn.Conv2d(64, 128, 3, 1, 1).cuda()
    ref.load_state_dict(m.state_dict())

    x = torch.randn(4, 64, 16, 16, device="cuda")
    y_ref = ref(x)
    y_hip = m(x.bfloat16().contiguous(memory_format=CL))
    assert y_hip.dtype == torch.bfloat16
    err = (y_hip.float() - y_ref).abs().max().item()
    assert err < 0.05 * y_ref.abs().max().item() + 0.05


def test_model_step_custom_convs_bf16():
    """Whole training step with the custom conv path under autocast."""
    import numpy as np

    from p2pvg_amd.core import Config
    from p2pvg_amd.models import P2PModel

    cfg = Config(dataset="bair", backbone="vgg", channels=3, batch_size=2,
                 max_seq_len=5, g_dim=64, z_dim=8, rnn_size=128, device="cuda",
                 skip_prob=0.0, dtype="bf16")
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg).to("cuda").to(memory_format=torch.channels_last)
    x = torch.rand(5, 2, 3, 64, 64, device="cuda")
    first = None
    for i in range(4):
        model.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            losses = model(x, 0, 4)
        if first is None:
            first = float(losses[0])
        last = float(losses[0])
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses)
    assert last < first
