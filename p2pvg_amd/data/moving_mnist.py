"""On-the-fly bouncing-MNIST synthesis with dynamic sequence length.

API- and semantics-compatible with the reference dataset
(reference data/moving_mnist.py:8-105): 32px digits on an image_size canvas,
velocities in [-4,4], stochastic wall bounce (non-deterministic mode redraws
velocity on contact), additive compositing clamped to 1, per-worker one-shot
numpy seeding, and `get_seq_len()` drawing U[max-2*delta, max].

The reference loads digits through torchvision's MNIST (unavailable offline
here); we read the raw idx files directly from data_root/MNIST/raw when
present, and otherwise synthesize procedural digit glyphs (PIL-rendered
characters) so the pipeline runs with no dataset on disk — benchmarks use that
synthetic path and say so.
"""
from __future__ import annotations

import gzip
import os
import struct
from typing import Optional

import numpy as np
import torch


def _read_idx_images(path: str) -> np.ndarray:
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        magic, n, rows, cols = struct.unpack(">IIII", f.read(16))
        assert magic == 2051, f"bad idx magic {magic} in {path}"
        data = np.frombuffer(f.read(n * rows * cols), dtype=np.uint8)
        return data.reshape(n, rows, cols)


def _find_mnist_images(data_root: str, train: bool) -> Optional[np.ndarray]:
    stem = "train-images-idx3-ubyte" if train else "t10k-images-idx3-ubyte"
    for sub in ("MNIST/raw", "mnist", "."):
        for suffix in ("", ".gz"):
            p = os.path.join(data_root, sub, stem + suffix)
            if os.path.exists(p):
                return _read_idx_images(p)
    return None


def _procedural_digits(n: int = 640, size: int = 28, seed: int = 1234) -> np.ndarray:
    """Render digit glyphs 0-9 with PIL at several jittered positions/scales.

    Gives MNIST-shaped (n,28,28) uint8 sprites with the same value range so the
    synthesis path and every downstream shape is identical to the real-data path.
    """
    from PIL import Image, ImageDraw, ImageFont

    rng = np.random.RandomState(seed)
    font = ImageFont.load_default()
    out = np.zeros((n, size, size), dtype=np.uint8)
    for i in range(n):
        d = str(i % 10)
        img = Image.new("L", (size, size), 0)
        draw = ImageDraw.Draw(img)
        # draw small, then rescale up with jitter for variety
        small = Image.new("L", (10, 12), 0)
        ImageDraw.Draw(small).text((2, 0), d, fill=255, font=font)
        scale = 1.4 + rng.uniform(0, 0.8)
        w, h = int(10 * scale), int(12 * scale)
        big = small.resize((w, h), Image.BILINEAR)
        ox = rng.randint(0, max(1, size - w))
        oy = rng.randint(0, max(1, size - h))
        img.paste(big, (ox, oy))
        out[i] = np.asarray(img, dtype=np.uint8)
    return out


class DynamicLengthMovingMNIST(torch.utils.data.Dataset):
    """Bouncing MNIST generated on the fly (reference data/moving_mnist.py:8)."""

    def __init__(
        self,
        data_root: str = "data_root",
        train: bool = True,
        transform=None,
        max_seq_len: int = 20,
        n_past: int = 1,
        delta_len: int = 3,
        image_size: int = 64,
        num_digits: int = 2,
        deterministic: bool = True,
        opt=None,
    ):
        self.train = train
        self.max_seq_len = max_seq_len
        self.n_past = n_past
        self.delta_len = delta_len
        self.image_size = image_size
        self.num_digits = num_digits
        self.deterministic = deterministic
        self.seed_is_set = False  # multi-process loading: one-shot seed
        self.channels = 1
        self.digit_size = 32

        imgs = _find_mnist_images(data_root, train)
        self.synthetic = imgs is None
        if imgs is None:
            imgs = _procedural_digits(640 if train else 160)
        # pre-scale all digits to 32px once (the reference re-scales per access
        # through a torchvision transform; doing it once is pure win)
        t = torch.from_numpy(imgs.astype(np.float32) / 255.0)
        self.digits = torch.nn.functional.interpolate(
            t.unsqueeze(1), size=(self.digit_size, self.digit_size),
            mode="bilinear", align_corners=False,
        ).squeeze(1)
        self.N = self.digits.shape[0]

    def set_seed(self, seed: int) -> None:
        if not self.seed_is_set:
            self.seed_is_set = True
            np.random.seed(seed)

    def get_seq_len(self) -> int:
        return int(
            np.random.randint(
                low=self.max_seq_len - self.delta_len * 2, high=self.max_seq_len + 1
            )
        )

    def __len__(self) -> int:
        return self.N

    def __getitem__(self, index: int) -> torch.Tensor:
        self.set_seed(index)
        image_size = self.image_size
        digit_size = self.digit_size

        x = torch.zeros(self.max_seq_len, self.channels, image_size, image_size)

        for _ in range(self.num_digits):
            idx = np.random.randint(self.N)
            digit = self.digits[idx]

            sx = np.random.randint(image_size - digit_size)
            sy = np.random.randint(image_size - digit_size)
            dx = np.random.randint(-4, 5)
            dy = np.random.randint(-4, 5)
            for t in range(self.max_seq_len):
                if sy < 0:
                    sy = 0
                    if self.deterministic:
                        dy = -dy
                    else:
                        dy = np.random.randint(1, 5)
                        dx = np.random.randint(-4, 5)
                elif sy >= image_size - 32:
                    sy = image_size - 32 - 1
                    if self.deterministic:
                        dy = -dy
                    else:
                        dy = np.random.randint(-4, 0)
                        dx = np.random.randint(-4, 5)

                if sx < 0:
                    sx = 0
                    if self.deterministic:
                        dx = -dx
                    else:
                        dx = np.random.randint(1, 5)
                        dy = np.random.randint(-4, 5)
                elif sx >= image_size - 32:
                    sx = image_size - 32 - 1
                    if self.deterministic:
                        dx = -dx
                    else:
                        dx = np.random.randint(-4, 0)
                        dy = np.random.randint(-4, 5)

                x[t, 0, sy : sy + 32, sx : sx + 32] += digit
                sy += dy
                sx += dx

        x.clamp_(max=1.0)
        return x
