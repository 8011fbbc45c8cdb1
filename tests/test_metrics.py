"""SSIM/PSNR metric tests against analytic expectations."""
import torch

from p2pvg_amd.utils.metrics import end_frame_ssim, mse, psnr, ssim


def test_ssim_identity():
    x = torch.rand(2, 3, 64, 64)
    s = ssim(x, x)
    assert torch.allclose(s, torch.ones(2), atol=1e-4)


def test_ssim_decreases_with_noise():
    torch.manual_seed(0)
    x = torch.rand(2, 1, 64, 64)
    s_small = ssim(x, (x + 0.05 * torch.randn_like(x)).clamp(0, 1)).mean()
    s_big = ssim(x, (x + 0.5 * torch.randn_like(x)).clamp(0, 1)).mean()
    assert s_small > s_big
    assert 0 <= s_big < s_small <= 1.0001


def test_psnr_known_value():
    x = torch.zeros(1, 1, 16, 16)
    y = torch.full_like(x, 0.1)
    # mse = 0.01 -> psnr = 10*log10(1/0.01) = 20
    assert torch.allclose(psnr(x, y), torch.tensor([20.0]), atol=1e-4)


def test_end_frame_ssim_list_input():
    x = torch.rand(2, 1, 32, 32)
    gen = [torch.rand(2, 1, 32, 32) for _ in range(5)] + [x.clone()]
    s = end_frame_ssim(gen, x)
    assert torch.allclose(s, torch.ones(2), atol=1e-4)


def test_gaussian_kl_closed_form_cpu():
    """ops.gaussian_kl matches the analytic KL of two diagonal Gaussians and
    normalizes by the CONFIGURED denom (reference misc/criterion.py:10-15
    divides by opt.batch_size, not the runtime batch)."""
    import math

    import torch

    from p2pvg_amd import ops

    # identical distributions -> zero
    mu = torch.randn(4, 3)
    lv = torch.randn(4, 3)
    z = ops.gaussian_kl(mu, lv, mu.clone(), lv.clone(), 4.0)
    assert abs(float(z)) < 1e-6

    # hand value: KL(N(1, e^0) || N(0, e^0)) = 0.5 per element
    mu1 = torch.ones(2, 5)
    zero = torch.zeros(2, 5)
    v = ops.gaussian_kl(mu1, zero, zero, zero, 10.0)
    assert abs(float(v) - 0.5 * 10 / 10.0) < 1e-6

    # general case vs torch.distributions, denom != runtime batch
    torch.manual_seed(0)
    m1, l1 = torch.randn(3, 7), torch.randn(3, 7)
    m2, l2 = torch.randn(3, 7), torch.randn(3, 7)
    ref = torch.distributions.kl_divergence(
        torch.distributions.Normal(m1, (0.5 * l1).exp()),
        torch.distributions.Normal(m2, (0.5 * l2).exp()),
    ).sum() / 13.0
    got = ops.gaussian_kl(m1, l1, m2, l2, 13.0)
    assert torch.allclose(got, ref, atol=1e-5), (float(got), float(ref))
