"""Dataset factory and infinite batch generators.

Capability parity with reference data/data_utils.py:6-140: `load_dataset(cfg)`
dispatches on cfg.dataset; `get_data_generator` yields (t, b, ...) batches on
the target device, truncated to a per-batch dynamic length drawn by the
dataset. MI355X-native differences: pinned host memory + non_blocking HtoD
(the reference does a blocking `.cuda()` per batch, data_utils.py:100,116),
device is configurable (CPU works), and under DDP each rank seeds its workers
with a rank offset so shards differ.
"""
from __future__ import annotations

import os
from typing import Iterator, Tuple

import torch
from torch.utils.data import DataLoader

from .bair import BairRobotPush
from .human36m import Human36mDataset
from .moving_mnist import DynamicLengthMovingMNIST
from .weizmann import WeizmannDataset


def load_dataset(cfg, eval=False, eval_len=None):
    if cfg.dataset == "mnist":
        mk = lambda train: DynamicLengthMovingMNIST(  # noqa: E731
            train=train,
            data_root=cfg.data_root,
            max_seq_len=cfg.max_seq_len,
            delta_len=cfg.delta_len,
            image_size=cfg.image_width,
            deterministic=False,
            num_digits=cfg.num_digits,
        )
        return mk(True), mk(False)

    if cfg.dataset == "weizmann":
        assert cfg.channels == 3, f"weizmann has 3 channels, got {cfg.channels}"
        mk = lambda train, msl: WeizmannDataset(  # noqa: E731
            data_root=cfg.data_root,
            train=train,
            max_seq_len=msl,
            n_past=cfg.n_past,
            delta_len=cfg.delta_len,
            image_size=cfg.image_width,
        )
        return mk(True, 18), mk(False, 10)

    if cfg.dataset == "h36m":
        data_root = os.path.join(cfg.data_root, "processed/h36m-fetch/processed")
        mk = lambda mode, spd: Human36mDataset(  # noqa: E731
            data_root=data_root,
            max_seq_len=30,
            delta_len=cfg.delta_len,
            speed_range=spd,
            n_breakpoints=0,
            acc_range=[0, 0],
            mode=mode,
        )
        return mk("train", [6, 6]), mk("test", [1, 1])

    if cfg.dataset == "bair":
        assert cfg.channels == 3, f"bair has 3 channels, got {cfg.channels}"
        mk = lambda train: BairRobotPush(  # noqa: E731
            data_root=cfg.data_root,
            train=train,
            max_seq_len=cfg.max_seq_len,
            delta_len=cfg.delta_len,
            image_size=cfg.image_width,
        )
        return mk(True), mk(False)

    raise ValueError(f"Unknown dataset {cfg.dataset!r}")


def _worker_seed_fn(rank: int):
    def _init(worker_id: int):
        import numpy as np

        seed = (torch.initial_seed() + rank * 7919 + worker_id) % (2**31)
        np.random.seed(seed)

    return _init


def _make_loader(data, batch_size: int, cfg, rank: int = 0) -> DataLoader:
    pin = torch.cuda.is_available()
    return DataLoader(
        data,
        batch_size=batch_size,
        shuffle=True,
        drop_last=True,
        num_workers=getattr(cfg, "num_workers", 1),
        pin_memory=pin,
        worker_init_fn=_worker_seed_fn(rank),
        persistent_workers=getattr(cfg, "num_workers", 1) > 0,
    )


def get_generator(loader: DataLoader, device, dynamic_length: bool = True) -> Iterator:
    cuda = getattr(device, "type", str(device)).startswith("cuda")
    while True:
        for data in loader:
            if cuda:
                # transfer the PINNED collated batch as-is (a host-side
                # permute().contiguous() would allocate an unpinned
                # intermediate and silently degrade the copy to synchronous),
                # then do the (b,t)->(t,b) transpose on device where it costs
                # one ~30us pass at HBM bandwidth. Store the clip NHWC so
                # each x[i] is a channels_last (B,C,H,W) frame — the NHWC
                # conv path then never re-lays frames out per encoder call.
                data = data.to(device, non_blocking=True)
                data = (data.permute(1, 0, 3, 4, 2).contiguous()
                        .permute(0, 1, 4, 2, 3))
            else:
                data = data.permute(1, 0, 2, 3, 4).contiguous().to(device)
            if dynamic_length:
                data = data[: loader.dataset.get_seq_len()]
            yield data


def get_h36m_generator(loader: DataLoader, device, dynamic_length: bool = True) -> Iterator:
    cuda = getattr(device, "type", str(device)).startswith("cuda")
    while True:
        for data in loader:
            seq_len = loader.dataset.get_seq_len()
            if cuda:
                # pinned batch HtoD first, permute/cast on device (see above)
                pose_2d = data["pose_2d"].to(device, non_blocking=True).permute(1, 0, 2, 3).float()
                pose_3d = data["pose_3d"].to(device, non_blocking=True).permute(1, 0, 2, 3).float()
            else:
                pose_2d = data["pose_2d"].permute(1, 0, 2, 3).float().to(device)
                pose_3d = data["pose_3d"].permute(1, 0, 2, 3).float().to(device)
            camera_view = data["camera_view"]
            if dynamic_length:
                pose_2d = pose_2d[:seq_len]
                pose_3d = pose_3d[:seq_len]
            yield (pose_2d, pose_3d, camera_view)


def get_data_generator(data, train: bool = True, dynamic_length: bool = True, opt=None, rank: int = 0):
    cfg = opt
    device = torch.device(cfg.resolved_device() if hasattr(cfg, "resolved_device") else "cpu")
    if cfg.dataset == "h36m":
        bs = cfg.batch_size if train else 10
        loader = _make_loader(data, bs, cfg, rank)
        return get_h36m_generator(loader, device, dynamic_length)
    loader = _make_loader(data, cfg.batch_size, cfg, rank)
    return get_generator(loader, device, dynamic_length)
