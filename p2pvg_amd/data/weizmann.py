"""Weizmann action dataset: all frames RAM-resident at init, h-flip augmented.

Capability parity with reference data/weizmann.py:11-117: directory layout
data_root/weizmann/<identity>/<action>/<frame>.png; train/test = first 2/3 vs
last 1/3 of each clip; every clip stored twice (original + horizontal flip);
random temporal crop of max_seq_len per access; dynamic length U[10,18] train /
U[6,10... max] test.

Falls back to synthetic clips when the directory is absent (offline plumbing).
"""
from __future__ import annotations

import os

import numpy as np
import torch


class WeizmannDataset(torch.utils.data.Dataset):
    def __init__(
        self,
        data_root: str = "data_root",
        train: bool = True,
        transform=None,
        max_seq_len: int = 18,
        n_past: int = 1,
        delta_len: int = 3,
        image_size: int = 64,
        opt=None,
        synthetic: bool = False,
    ):
        self.root = os.path.join(data_root, "weizmann")
        self.train = train
        self.max_seq_len = max_seq_len
        self.n_past = n_past
        self.delta_len = delta_len
        self.channels = 3
        self.image_size = image_size
        self.seed_is_set = False

        self.data = []
        if os.path.isdir(self.root) and not synthetic:
            self._load_clips()
        self.synthetic = synthetic or not self.data
        if self.synthetic:
            self._synth_len = 64

    def _load_clips(self):
        from PIL import Image

        s = self.image_size
        ids = [
            d for d in sorted(os.listdir(self.root))
            if os.path.isdir(os.path.join(self.root, d))
        ]
        for identity in ids:
            for act in sorted(os.listdir(os.path.join(self.root, identity))):
                adir = os.path.join(self.root, identity, act)
                frames = sorted(os.listdir(adir))
                num_train = len(frames) * 2 // 3
                start, end = (0, num_train) if self.train else (num_train, len(frames))
                n_frames = end - start
                if n_frames < self.max_seq_len:
                    continue
                seq = torch.zeros(n_frames, 3, s, s)
                seq_flip = torch.zeros(n_frames, 3, s, s)
                for t in range(start, end):
                    with Image.open(os.path.join(adir, frames[t])) as im:
                        arr = np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0
                    ten = torch.from_numpy(arr).permute(2, 0, 1)
                    seq[t - start] = ten
                    seq_flip[t - start] = torch.flip(ten, dims=[2])
                self.data.append({"seq": seq, "n_frames": n_frames})
                self.data.append({"seq": seq_flip, "n_frames": n_frames})

    def set_seed(self, seed: int) -> None:
        if not self.seed_is_set:
            self.seed_is_set = True
            np.random.seed(seed)

    def get_seq_len(self) -> int:
        # reference data/weizmann.py:95-101: U[10,18] train, U[6,max] test
        if self.train:
            return int(np.random.randint(low=10, high=self.max_seq_len + 1))
        return int(np.random.randint(low=6, high=self.max_seq_len + 1))

    def __len__(self) -> int:
        return self._synth_len if self.synthetic else len(self.data)

    def __getitem__(self, idx: int) -> torch.Tensor:
        self.set_seed(idx)
        s = self.image_size
        if self.synthetic:
            base = torch.rand(3, s, s)
            drift = torch.randn(self.max_seq_len, 3, 1, 1) * 0.05
            return (base.unsqueeze(0) + drift.cumsum(0)).clamp_(0, 1)

        data = self.data[idx]
        seq, n_frames = data["seq"], data["n_frames"]
        start_ix = np.random.randint(low=0, high=n_frames - self.max_seq_len + 1)
        return seq[start_ix : start_ix + self.max_seq_len]
