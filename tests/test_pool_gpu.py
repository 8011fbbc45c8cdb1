"""GPU tests for NHWC maxpool 2x2 / nearest-x2 upsample kernels vs ATen."""
import pytest
import torch

pytestmark = pytest.mark.gpu

CL = torch.channels_last


@pytest.mark.parametrize("N,C,H,W", [(4, 64, 32, 32), (2, 128, 16, 16), (3, 8, 64, 64)])
def test_maxpool_fwd_bwd(N, C, H, W):
    from p2pvg_amd.ops.pool import MaxPool2x2Fn

    torch.manual_seed(0)
    x = torch.randn(N, C, H, W, device="cuda")
    xr = x.clone().requires_grad_()
    y_ref = torch.nn.functional.max_pool2d(xr, 2, 2)
    g = torch.randn_like(y_ref)
    y_ref.backward(g)

    xh = x.bfloat16().contiguous(memory_format=CL).requires_grad_()
    y = MaxPool2x2Fn.apply(xh)
    y.backward(g.bfloat16().contiguous(memory_format=CL))

    assert torch.allclose(y.float(), y_ref, rtol=1e-2, atol=1e-2)
    # bf16 rounding can flip argmax for near-ties; compare grads loosely
    diff = (xh.grad.float() - xr.grad).abs()
    assert (diff > 0.05).float().mean().item() < 0.01


@pytest.mark.parametrize("N,C,H,W", [(4, 64, 8, 8), (2, 512, 4, 4), (3, 8, 32, 32)])
def test_upsample_fwd_bwd(N, C, H, W):
    from p2pvg_amd.ops.pool import Upsample2xFn

    torch.manual_seed(1)
    x = torch.randn(N, C, H, W, device="cuda")
    xr = x.clone().requires_grad_()
    y_ref = torch.nn.functional.interpolate(xr, scale_factor=2, mode="nearest")
    g = torch.randn_like(y_ref)
    y_ref.backward(g)

    xh = x.bfloat16().contiguous(memory_format=CL).requires_grad_()
    y = Upsample2xFn.apply(xh)
    y.backward(g.bfloat16().contiguous(memory_format=CL))

    assert torch.allclose(y.float(), y_ref, rtol=1e-2, atol=1e-2)
    assert torch.allclose(xh.grad.float(), xr.grad, rtol=5e-2, atol=5e-2)
