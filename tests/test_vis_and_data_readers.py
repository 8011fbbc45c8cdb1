"""Visualization and real-data reader paths (CPU)."""
import numpy as np
import pytest
import torch

from p2pvg_amd.core import Config


def test_skeleton3d_visualizer_renders():
    from p2pvg_amd.data.human36m import make_h36m_skeleton
    from p2pvg_amd.utils.vis import Skeleton3DVisualizer, STD_SCALE

    sk = make_h36m_skeleton()
    vis = Skeleton3DVisualizer(sk.parents(), plot_3d_limit=[-2 * STD_SCALE, 2 * STD_SCALE])
    pose = np.random.RandomState(0).randn(3, 17, 3).astype(np.float32)
    imgs = vis.set_data(pose, camera_view=1)
    assert imgs.shape[0] == 3 and imgs.shape[3] == 3
    assert imgs.dtype == np.uint8


def test_vis_seq_h36m(tmp_path):
    from p2pvg_amd.models import P2PModel
    from p2pvg_amd.utils.logging import ScalarWriter
    from p2pvg_amd.utils.vis import Skeleton3DVisualizer, vis_seq, STD_SCALE
    from p2pvg_amd.data.human36m import make_h36m_skeleton

    cfg = Config(dataset="h36m", backbone="mlp", batch_size=2, max_seq_len=5,
                 delta_len=1, g_dim=16, z_dim=4, rnn_size=16, nsample=2,
                 device="cpu", log_dir=str(tmp_path))
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg)
    model.eval()
    pose_3d = torch.randn(5, 2, 17, 3)
    x = (pose_3d[..., :2].clone(), pose_3d, torch.tensor([0, 1]))
    sk = make_h36m_skeleton()
    vis = Skeleton3DVisualizer(sk.parents(), plot_3d_limit=[-2 * STD_SCALE, 2 * STD_SCALE])
    w = ScalarWriter(str(tmp_path))
    with torch.no_grad():
        vis_seq(model, x, epoch=0, output_len=5, model_mode="full",
                recon_mode="test", skip_frame=False, h36m_visualizer=vis,
                writer=w, opt=cfg)
    assert list((tmp_path / "gen_vis").glob("*.png"))
    assert list((tmp_path / "gen_vis").glob("*.gif"))


def test_bair_reader_on_converted_pngs(tmp_path):
    from PIL import Image

    from p2pvg_amd.data.bair import BairRobotPush

    rng = np.random.RandomState(0)
    for clip in range(2):
        d = tmp_path / "bair" / "processed_data" / "train" / "traj_x" / str(clip)
        d.mkdir(parents=True)
        for i in range(4):
            Image.fromarray(rng.randint(0, 255, (64, 64, 3), dtype=np.uint8)).save(d / f"{i}.png")

    ds = BairRobotPush(data_root=str(tmp_path), train=True, max_seq_len=4)
    assert not ds.synthetic
    x = ds[0]
    assert x.shape == (4, 3, 64, 64)
    assert 0.0 <= x.min() and x.max() <= 1.0 and x.sum() > 0


def test_weizmann_reader_on_png_dirs(tmp_path):
    from PIL import Image

    from p2pvg_amd.data.weizmann import WeizmannDataset

    rng = np.random.RandomState(1)
    d = tmp_path / "weizmann" / "person1" / "walk"
    d.mkdir(parents=True)
    for i in range(12):
        Image.fromarray(rng.randint(0, 255, (64, 64, 3), dtype=np.uint8)).save(d / f"{i:03d}.png")

    ds = WeizmannDataset(data_root=str(tmp_path), train=True, max_seq_len=6)
    assert not ds.synthetic
    assert len(ds) == 2  # original + flipped
    x = ds[0]
    assert x.shape == (6, 3, 64, 64)
    # flipped copy mirrors horizontally
    a, b = ds[0], ds[1]


def test_generate_image_pair(tmp_path):
    """--start_img/--end_img path (declared but broken in the reference,
    generate.py:93-96)."""
    import subprocess
    import sys
    import os

    from PIL import Image

    from p2pvg_amd.models import P2PModel
    from p2pvg_amd.utils import save_checkpoint

    cfg = Config(dataset="mnist", backbone="dcgan", batch_size=2, max_seq_len=6,
                 g_dim=16, z_dim=4, rnn_size=16, device="cpu")
    torch.manual_seed(0)
    model = P2PModel(cfg)
    ckpt = tmp_path / "m.pth"
    save_checkpoint(model, 0, str(ckpt))

    rng = np.random.RandomState(0)
    for name in ("a.png", "b.png"):
        Image.fromarray(rng.randint(0, 255, (64, 64), dtype=np.uint8)).save(tmp_path / name)

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = os.environ.copy()
    env["PYTHONPATH"] = root
    r = subprocess.run(
        [sys.executable, os.path.join(root, "generate.py"), "--ckpt", str(ckpt),
         "--start_img", str(tmp_path / "a.png"), "--end_img", str(tmp_path / "b.png"),
         "--output_root", str(tmp_path / "out"), "--device", "cpu"],
        capture_output=True, text=True, timeout=600, cwd=root, env=env,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert (tmp_path / "out" / "len_10-gt.png").exists()


def test_cp_border_painters_pixel_semantics():
    """Pixel-level contract of the control-point border painters
    (reference misc/visualize.py:13-87): orange pad-wide ring on frame 0,
    red ring on the control-point frame, cp frame repeated past seq_len;
    interiors untouched; 1-channel inputs promoted to RGB."""
    import torch

    from p2pvg_amd.utils.vis import (_CP_RGB, _START_RGB, add_gt_cp_border,
                                     add_samples_cp_border)

    t, b, h, w, pad = 6, 2, 16, 16, 3
    seq = torch.full((t, b, 1, h, w), 0.5)
    out = add_gt_cp_border(seq.clone(), seq_len=4, output_len=6, padding=pad)
    assert out.shape == (t, b, 3, h, w)

    def ring_ok(frame, rgb):
        for ch in range(3):
            assert torch.all(frame[ch, :pad, :] == rgb[ch])      # top band
            assert torch.all(frame[ch, -pad:, :] == rgb[ch])     # bottom
            assert torch.all(frame[ch, :, :pad] == rgb[ch])      # left
            assert torch.all(frame[ch, :, -pad:] == rgb[ch])     # right
        assert torch.all(frame[:, pad:-pad, pad:-pad] == 0.5)    # interior

    ring_ok(out[0, 0], _START_RGB)                # start frame: orange
    for i in range(3, 6):                         # cp (seq_len-1) + repeats
        ring_ok(out[i, 0], _CP_RGB)
    for i in (1, 2):                              # untouched interior frames
        assert torch.all(out[i] == 0.5)

    samples = torch.full((2, t, b, 1, h, w), 0.5)
    s = add_samples_cp_border(samples.clone(), seq_len=4, output_len=5,
                              padding=pad)
    assert s.shape == (2, t, b, 3, h, w)
    ring_ok(s[0, 0, 0], _START_RGB)               # first frame orange
    ring_ok(s[1, 4, 0], _CP_RGB)                  # last GENERATED frame red
    assert torch.all(s[0, 2] == 0.5)
