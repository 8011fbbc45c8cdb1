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
