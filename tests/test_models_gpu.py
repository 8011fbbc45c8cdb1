"""GPU model-family coverage: every backbone trains a step on the custom
kernel path (bf16, channels_last) with finite losses."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from p2pvg_amd.core import Config
from p2pvg_amd.models import P2PModel


@pytest.mark.parametrize("backbone,width,channels,dataset", [
    ("dcgan", 64, 1, "mnist"),
    ("dcgan", 128, 3, "bair"),
    ("vgg", 64, 3, "bair"),
    ("vgg", 128, 3, "bair"),
])
def test_backbone_family_trains_on_gpu(backbone, width, channels, dataset):
    cfg = Config(dataset=dataset, backbone=backbone, image_width=width,
                 channels=channels, batch_size=2, max_seq_len=4, g_dim=64,
                 z_dim=8, rnn_size=64, device="cuda", skip_prob=0.0,
                 dtype="bf16")
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg).to("cuda").to(memory_format=torch.channels_last)
    x = torch.rand(4, 2, channels, width, width, device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        losses = model(x, 0, 3)
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses), losses

    # generation path (eval BN + fused kernels)
    model.eval()
    with torch.no_grad():
        out = model.p2p_generate(x, 5, 4, model_mode="full")
    torch.cuda.synchronize()
    assert len(out) == 5
    assert all(torch.isfinite(f.float()).all() for f in out)


def test_h36m_mlp_trains_on_gpu():
    cfg = Config(dataset="h36m", backbone="mlp", batch_size=4, max_seq_len=6,
                 g_dim=64, z_dim=8, rnn_size=128, device="cuda",
                 skip_prob=0.0, dtype="bf16")
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg).to("cuda")
    pose_3d = torch.randn(6, 4, 17, 3, device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        losses = model((pose_3d[..., :2], pose_3d, [0] * 4), 0, 5)
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses)
