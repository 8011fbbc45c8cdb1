"""Human3.6M skeleton dataset (non-image modality path).

Capability parity with reference data/human36m/human36m.py:26-287:
- reads annot.h5 per (subject, action) from <data_root>/<S*>/<action>/annot.h5
  (h36m-fetch layout); train subjects S1/5/6/7/8, test S9/11;
- 32 -> 17 joint reduction + shoulder re-wiring (parents[11]=parents[14]=8);
- whole-dataset standardization of 2d/3d poses to N(0, STD_SCALE^2);
- subsequence crop with speed / breakpoint machinery;
- returns dict {pose_2d, pose_3d, camera_view, speed, breakpoints};
- dynamic length via get_seq_len() U[max-2*delta, max].

h5py is optional offline: when absent (or the directory is missing) the
dataset synthesizes smooth random 17x3 skeleton sequences of the same shape
and statistics so the h36m_mlp model path runs end-to-end (data=synthetic).
"""
from __future__ import annotations

import os
from typing import Dict, List

import numpy as np
import torch

from .skeleton import Skeleton

STD_SCALE = 3

H36M_PARENTS_32 = [
    -1, 0, 1, 2, 3, 4, 0, 6, 7, 8, 9, 0, 11, 12, 13, 14, 12,
    16, 17, 18, 19, 20, 19, 22, 12, 24, 25, 26, 27, 28, 27, 30,
]
H36M_JOINTS_LEFT = [6, 7, 8, 9, 10, 16, 17, 18, 19, 20, 21, 22, 23]
H36M_JOINTS_RIGHT = [1, 2, 3, 4, 5, 24, 25, 26, 27, 28, 29, 30, 31]
H36M_STATIC_JOINTS = [4, 5, 9, 10, 11, 16, 20, 21, 22, 23, 24, 28, 29, 30, 31]


def make_h36m_skeleton(remove_static_joints: bool = True) -> Skeleton:
    sk = Skeleton(H36M_PARENTS_32, H36M_JOINTS_LEFT, H36M_JOINTS_RIGHT)
    if remove_static_joints:
        sk.remove_joints(H36M_STATIC_JOINTS)
        # re-wire shoulders to the thorax (reference human36m.py:61-62)
        sk._parents[11] = 8
        sk._parents[14] = 8
    return sk


def read_human36m(root_dir: str, mode: str = "train") -> Dict:
    """Read annot.h5 files (reference data/human36m/human36m.py:123-165)."""
    import h5py

    mode_id = ["S1", "S5", "S6", "S7", "S8"] if mode == "train" else ["S9", "S11"]
    data_dict: Dict = {"annot": []}
    for id_n in sorted(os.listdir(root_dir)):
        if id_n not in mode_id:
            continue
        for act_n in sorted(os.listdir(os.path.join(root_dir, id_n))):
            annot_path = os.path.join(root_dir, id_n, act_n, "annot.h5")
            if not os.path.exists(annot_path):
                continue
            with h5py.File(annot_path, "r") as h5:
                annot = {"path": annot_path}
                for k, v in h5.items():
                    if isinstance(v, type(h5)) or hasattr(v, "items"):
                        annot[k] = {k_: np.array(v_) for k_, v_ in v.items()}
                    else:
                        annot[k] = np.array(v)
                data_dict["annot"].append(annot)
    return data_dict


def reformat_data(raw_data: Dict) -> Dict:
    """The h5 stacks 4 camera views along time; keep view 0 per sequence
    (reference human36m.py:168-183 uses get_1view_data)."""
    data: Dict = {"pose": {"2d": [], "3d": []}, "camera_view": []}
    for annot in raw_data["annot"]:
        n = annot["pose"]["2d"].shape[0] // 4
        data["pose"]["2d"].append(annot["pose"]["2d"][:n])
        data["pose"]["3d"].append(annot["pose"]["3d"][:n])
        data["camera_view"].extend([0, 1, 2, 3])
    return data


def standardize_dataset(data: Dict, scale: float = STD_SCALE) -> None:
    """Whole-dataset standardization to N(0, scale^2)
    (reference align_and_normalize_dataset_v2, human36m.py:233-270)."""
    p2, p3 = data["pose"]["2d"], data["pose"]["3d"]
    total = sum(s.shape[0] * s.shape[1] for s in p2)
    if total == 0:
        return
    mean2 = sum(s.sum(axis=(0, 1)) for s in p2) / total
    mean3 = sum(s.sum(axis=(0, 1)) for s in p3) / total
    std2 = np.sqrt(sum(((s - mean2) ** 2).sum(axis=(0, 1)) for s in p2) / total)
    std3 = np.sqrt(sum(((s - mean3) ** 2).sum(axis=(0, 1)) for s in p3) / total)
    for i in range(len(p2)):
        p2[i] = scale * (p2[i] - mean2) / std2
        p3[i] = scale * (p3[i] - mean3) / std3


def _synthetic_sequences(n_seq: int, length: int, seed: int = 7):
    """Smooth random walks shaped like standardized 17-joint poses."""
    rng = np.random.RandomState(seed)
    seqs3, seqs2 = [], []
    for _ in range(n_seq):
        base3 = rng.randn(17, 3)
        steps = rng.randn(length, 17, 3) * 0.08
        seq3 = (base3[None] + np.cumsum(steps, axis=0)).astype(np.float32)
        seqs3.append(seq3)
        seqs2.append(seq3[:, :, :2].copy())
    return seqs2, seqs3


class Human36mDataset(torch.utils.data.Dataset):
    def __init__(
        self,
        data_root: str,
        max_seq_len: int,
        delta_len: int,
        n_breakpoints: int = 0,
        speed_range=(1, 1),
        acc_range=(-1, 1),
        train: bool = True,
        remove_static_joints: bool = True,
        mode: str = "train",
        synthetic: bool = False,
    ):
        assert mode in ("train", "test")
        self.data_root = os.path.abspath(os.path.expanduser(data_root))
        self.max_seq_len = max_seq_len
        self.delta_len = delta_len
        self.speed_range = list(speed_range)
        self.n_breakpoints = n_breakpoints
        self.acc_range = list(acc_range)
        self.train = train
        self.mode = mode
        self.skeleton = make_h36m_skeleton(remove_static_joints=False)

        have_h5 = True
        try:
            import h5py  # noqa: F401
        except ImportError:
            have_h5 = False

        self.synthetic = synthetic or not have_h5 or not os.path.isdir(self.data_root)
        if not self.synthetic:
            raw = read_human36m(self.data_root, self.mode)
            self.data = reformat_data(raw)
            self.data["pose"]["3d"] = [
                s for s in self.data["pose"]["3d"] if s.shape[0] >= self.max_seq_len
            ]
            self.data["pose"]["2d"] = [
                s for s in self.data["pose"]["2d"] if s.shape[0] >= self.max_seq_len
            ]
            if remove_static_joints:
                kept = self.skeleton.remove_joints(H36M_STATIC_JOINTS)
                self.skeleton._parents[11] = 8
                self.skeleton._parents[14] = 8
                for i in range(len(self.data["pose"]["3d"])):
                    self.data["pose"]["3d"][i] = self.data["pose"]["3d"][i][:, kept]
                    self.data["pose"]["2d"][i] = self.data["pose"]["2d"][i][:, kept]
            standardize_dataset(self.data)
        else:
            if remove_static_joints:
                self.skeleton = make_h36m_skeleton(True)
            length = max(200, self.max_seq_len * max(self.speed_range) + 1)
            s2, s3 = _synthetic_sequences(64, length)
            self.data = {
                "pose": {"2d": s2, "3d": s3},
                "camera_view": list(np.tile([0, 1, 2, 3], 16)),
            }

    def get_seq_len(self) -> int:
        return int(
            np.random.randint(
                low=self.max_seq_len - 2 * self.delta_len, high=self.max_seq_len + 1
            )
        )

    def __len__(self) -> int:
        return len(self.data["pose"]["3d"])

    def __getitem__(self, idx: int) -> Dict:
        pose_2d = self.data["pose"]["2d"][idx]
        pose_3d = self.data["pose"]["3d"][idx]
        camera_view = self.data["camera_view"][idx % len(self.data["camera_view"])]

        total_len = pose_3d.shape[0]
        hi = total_len - self.speed_range[1] * self.max_seq_len + 1
        start = np.random.randint(low=0, high=max(1, hi))
        p2_crop: List[np.ndarray] = []
        p3_crop: List[np.ndarray] = []
        speed: List[int] = []

        if self.n_breakpoints > 0:
            # varying speed with breakpoints (reference human36m.py:78-92)
            off = 5
            bps = [0, self.max_seq_len] + list(
                np.random.randint(
                    low=1 + off, high=self.max_seq_len - off, size=self.n_breakpoints
                )
            )
            bps = sorted(bps)
            speed.append(
                int(np.random.randint(self.speed_range[0], self.speed_range[1] + 1))
            )
            for bp_i, bp in enumerate(bps[1:]):
                end = start + (bp - bps[bp_i]) * speed[-1]
                p2_crop.append(pose_2d[start:end:speed[-1]].copy())
                p3_crop.append(pose_3d[start:end:speed[-1]].copy())
                nxt = min(
                    max(
                        speed[-1]
                        + np.random.randint(self.acc_range[0], self.acc_range[1] + 1),
                        1,
                    ),
                    self.speed_range[1],
                )
                speed.append(int(nxt))
                start = end
            speed = speed[:-1]
        else:
            bps = []
            sp = int(np.random.randint(self.speed_range[0], self.speed_range[1] + 1))
            speed = sp
            p2_crop.append(pose_2d[start : start + self.max_seq_len * sp : sp].copy())
            p3_crop.append(pose_3d[start : start + self.max_seq_len * sp : sp].copy())

        return {
            "pose_2d": np.concatenate(p2_crop, 0),
            "pose_3d": np.concatenate(p3_crop, 0),
            "camera_view": camera_view,
            "speed": speed,
            "breakpoints": bps,
        }
