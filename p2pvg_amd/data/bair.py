"""BAIR robot-push dataset: PNG frame-directory reader.

Capability parity with reference data/bair.py:11-75: frames live in
data_root/bair/processed_data/{train,test}/<d1>/<d2>/<i>.png (produced by the
offline converter, see p2pvg_amd/data/convert_bair.py). Train samples a random
clip directory, test scans in order; __len__ is the reference's fixed 10000;
dynamic length via get_seq_len() U[max-2*delta, max].

If the dataset directory is absent, `synthetic=True` generates random clips of
the right shape so plumbing/bench runs work offline (and report data=synthetic).
"""
from __future__ import annotations

import os

import numpy as np
import torch


class BairRobotPush(torch.utils.data.Dataset):
    def __init__(
        self,
        data_root: str,
        train: bool = True,
        transform=None,
        max_seq_len: int = 30,
        delta_len: int = 5,
        image_size: int = 64,
        opt=None,
        synthetic: bool = False,
    ):
        self.root_dir = os.path.join(data_root, "bair")
        self.train = train
        self.max_seq_len = max_seq_len
        self.delta_len = delta_len
        self.image_size = image_size
        self.seed_is_set = False
        self.channels = 3

        sub = "train" if train else "test"
        self.data_dir = os.path.join(self.root_dir, "processed_data", sub)
        self.ordered = not train
        self.dirs = []
        if os.path.isdir(self.data_dir):
            for d1 in sorted(os.listdir(self.data_dir)):
                p1 = os.path.join(self.data_dir, d1)
                if not os.path.isdir(p1):
                    continue
                for d2 in sorted(os.listdir(p1)):
                    self.dirs.append(os.path.join(p1, d2))
        self.synthetic = synthetic or not self.dirs
        self.d = 0

    def get_seq_len(self) -> int:
        return int(
            np.random.randint(
                low=self.max_seq_len - self.delta_len * 2, high=self.max_seq_len + 1
            )
        )

    def set_seed(self, seed: int) -> None:
        if not self.seed_is_set:
            self.seed_is_set = True
            np.random.seed(seed)

    def __len__(self) -> int:
        return 10000

    def _load_png(self, fname: str) -> torch.Tensor:
        from PIL import Image

        with Image.open(fname) as im:
            arr = np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0
        return torch.from_numpy(arr).permute(2, 0, 1)

    def get_seq(self) -> torch.Tensor:
        s = self.image_size
        if self.synthetic:
            # smooth random video of the right shape (offline plumbing path)
            base = torch.rand(3, s, s)
            drift = torch.randn(self.max_seq_len, 3, 1, 1) * 0.05
            return (base.unsqueeze(0) + drift.cumsum(0)).clamp_(0, 1)

        if self.ordered:
            d = self.dirs[self.d]
            self.d = 0 if self.d == len(self.dirs) - 1 else self.d + 1
        else:
            d = self.dirs[np.random.randint(len(self.dirs))]

        image_seq = torch.zeros(self.max_seq_len, 3, s, s)
        for i in range(self.max_seq_len):
            image_seq[i] = self._load_png(os.path.join(d, f"{i}.png"))
        return image_seq

    def __getitem__(self, index: int) -> torch.Tensor:
        self.set_seed(index)
        return self.get_seq()
