"""Data layer tests: synthesis semantics, generator layout, dynamic length."""
import numpy as np
import torch

from p2pvg_amd import data as data_utils
from p2pvg_amd.core import Config
from p2pvg_amd.data.moving_mnist import DynamicLengthMovingMNIST
from p2pvg_amd.data.human36m import Human36mDataset
from p2pvg_amd.data.skeleton import Skeleton
from p2pvg_amd.data.human36m import make_h36m_skeleton


def test_moving_mnist_shapes_and_clamp():
    ds = DynamicLengthMovingMNIST(
        data_root="/nonexistent", train=True, max_seq_len=10, delta_len=2,
        image_size=64, num_digits=2, deterministic=False,
    )
    x = ds[0]
    assert x.shape == (10, 1, 64, 64)
    assert x.max() <= 1.0 and x.min() >= 0.0
    assert x.sum() > 0  # digits actually composited


def test_moving_mnist_dynamic_length_range():
    ds = DynamicLengthMovingMNIST(
        data_root="/nonexistent", max_seq_len=30, delta_len=5, deterministic=False
    )
    np.random.seed(0)
    lengths = {ds.get_seq_len() for _ in range(200)}
    assert min(lengths) >= 20 and max(lengths) <= 30
    assert len(lengths) > 5


def test_generator_layout_tbchw():
    cfg = Config(dataset="mnist", batch_size=3, max_seq_len=8, delta_len=1,
                 data_root="/nonexistent", device="cpu", num_workers=0)
    train, test = data_utils.load_dataset(cfg)
    gen = data_utils.get_data_generator(train, train=True, opt=cfg)
    x = next(gen)
    assert x.dim() == 5
    t, b, c, h, w = x.shape
    assert b == 3 and c == 1 and h == 64 and w == 64
    assert 6 <= t <= 8


def test_bair_weizmann_synthetic():
    from p2pvg_amd.data.bair import BairRobotPush
    from p2pvg_amd.data.weizmann import WeizmannDataset

    b = BairRobotPush(data_root="/nonexistent", train=True, max_seq_len=6)
    assert b.synthetic
    x = b[0]
    assert x.shape == (6, 3, 64, 64)

    w = WeizmannDataset(data_root="/nonexistent", train=True, max_seq_len=6)
    assert w.synthetic
    x = w[0]
    assert x.shape == (6, 3, 64, 64)


def test_h36m_synthetic_and_generator():
    ds = Human36mDataset(
        data_root="/nonexistent", max_seq_len=10, delta_len=2, mode="train"
    )
    assert ds.synthetic
    item = ds[0]
    assert item["pose_3d"].shape == (10, 17, 3)
    assert item["pose_2d"].shape == (10, 17, 2)

    cfg = Config(dataset="h36m", batch_size=2, max_seq_len=10, delta_len=2,
                 device="cpu", num_workers=0, data_root="/nonexistent")
    train, test = data_utils.load_dataset(cfg)
    gen = data_utils.get_data_generator(train, train=True, opt=cfg)
    p2, p3, cam = next(gen)
    assert p3.dim() == 4 and p3.shape[1] == 2 and p3.shape[2:] == (17, 3)


def test_skeleton_joint_removal_matches_h36m():
    sk = make_h36m_skeleton(remove_static_joints=True)
    assert sk.num_joints() == 17
    # shoulders re-wired to thorax
    assert sk.parents()[11] == 8 and sk.parents()[14] == 8
    # root
    assert sk.parents()[0] == -1


def test_skeleton_removal_generic():
    sk = Skeleton(parents=[-1, 0, 1, 2, 1], joints_left=[3], joints_right=[4])
    kept = sk.remove_joints([2])
    assert kept == [0, 1, 3, 4]
    # joint 3 (old) had parent 2 -> re-wired to 1, new index of 1 is 1
    assert list(sk.parents()) == [-1, 0, 1, 1]


def _png(path, s=64, v=None):
    import numpy as np
    from PIL import Image

    arr = (np.random.rand(s, s, 3) * 255).astype("uint8") if v is None else v
    Image.fromarray(arr).save(path)


def test_bair_real_png_dirs(tmp_path):
    """Real on-disk BAIR path: processed_data/<split>/<d1>/<d2>/<i>.png
    (the layout the converter writes), ordered scan in test mode."""
    import numpy as np

    for split in ("train", "test"):
        for d1 in ("traj_0", "traj_1"):
            for d2 in ("0", "1"):
                d = tmp_path / "bair" / "processed_data" / split / d1 / d2
                d.mkdir(parents=True)
                for i in range(12):
                    _png(d / f"{i}.png")

    from p2pvg_amd.data.bair import BairRobotPush

    ds = BairRobotPush(data_root=str(tmp_path), train=True, max_seq_len=12,
                       delta_len=2)
    assert not ds.synthetic, "real PNG dirs not picked up"
    seq = ds[0]
    assert seq.shape == (12, 3, 64, 64)
    assert 0.0 <= float(seq.min()) and float(seq.max()) <= 1.0

    dte = BairRobotPush(data_root=str(tmp_path), train=False, max_seq_len=12,
                        delta_len=2)
    assert not dte.synthetic and dte.ordered
    a, b = dte[0], dte[1]
    assert a.shape == b.shape == (12, 3, 64, 64)
    assert 10 <= dte.get_seq_len() <= 12  # U[max-2*delta, max]


def test_weizmann_real_frame_tree(tmp_path):
    """Real Weizmann path: <root>/weizmann/<identity>/<action>/<frames>,
    first 2/3 train; h-flip augmentation doubles the clip count."""
    d = tmp_path / "weizmann" / "daria" / "walk"
    d.mkdir(parents=True)
    for i in range(30):
        _png(d / f"{i:04d}.png")

    from p2pvg_amd.data.weizmann import WeizmannDataset

    ds = WeizmannDataset(data_root=str(tmp_path), train=True, max_seq_len=10,
                         delta_len=2)
    assert not ds.synthetic
    assert len(ds) == 2  # clip + flipped copy
    seq = ds[0]
    assert seq.shape == (10, 3, 64, 64)
    import torch

    flip = ds[1]
    assert flip.shape == (10, 3, 64, 64)


def test_moving_mnist_real_idx_files(tmp_path):
    """Real-MNIST path: raw idx3 file under data_root/MNIST/raw is read and
    used as digit sprites instead of the procedural fallback."""
    import gzip
    import struct

    import numpy as np

    raw = tmp_path / "MNIST" / "raw"
    raw.mkdir(parents=True)
    n = 32
    imgs = (np.random.rand(n, 28, 28) * 255).astype("uint8")
    payload = struct.pack(">IIII", 2051, n, 28, 28) + imgs.tobytes()
    with gzip.open(raw / "train-images-idx3-ubyte.gz", "wb") as f:
        f.write(payload)
    (raw / "t10k-images-idx3-ubyte").write_bytes(payload)

    from p2pvg_amd.data.moving_mnist import (DynamicLengthMovingMNIST,
                                             _find_mnist_images)

    loaded = _find_mnist_images(str(tmp_path), train=True)
    assert loaded is not None and loaded.shape == (n, 28, 28)
    assert np.array_equal(loaded, imgs)

    ds = DynamicLengthMovingMNIST(
        data_root=str(tmp_path), train=False, max_seq_len=8, num_digits=1,
        image_size=64, deterministic=True,
    )
    seq = ds[0]
    assert tuple(seq.shape) == (8, 64, 64, 1) or tuple(seq.shape) == (8, 1, 64, 64)
