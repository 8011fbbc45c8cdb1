"""Qualitative evaluation: sample grids, control-point borders, GIFs.

Capability parity with reference misc/visualize.py:13-272 — `vis_seq` runs
nsample p2p generations per model mode, paints orange/red borders on the
start/control-point frames, writes a PNG grid + GIF and logs image/video to
the scalar writer. The Human3.6M 3D skeleton renderer matches
reference data/human36m/human36m.py:290-366 (matplotlib 3D lines -> RGB).
"""
from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import torch

from .image import make_grid, save_gif, save_image, to_uint8_hwc

STD_SCALE = 3


def _ring(frames: torch.Tensor, rgb, pad: int) -> torch.Tensor:
    """A pad-wide colored ring around each frame; frames (..., 3, h, w)."""
    ringed = frames.new_empty(frames.shape)
    for ch in range(3):
        ringed[..., ch, :, :] = rgb[ch]
    ringed[..., pad:-pad, pad:-pad] = frames[..., pad:-pad, pad:-pad]
    return ringed


# start frame = orange, control-point frame = red (the parity target is the
# reference's exact border pixels, misc/visualize.py:13-87)
_START_RGB = (1.0, 165.0 / 255.0, 0.0)
_CP_RGB = (1.0, 0.0, 0.0)


def add_gt_cp_border(seq: torch.Tensor, seq_len: int, output_len: int, padding: int = 3):
    """Mark the ground-truth row: orange ring on frame 0, red ring on the
    control-point frame (and on its repeats past seq_len). seq: (t,b,c,h,w)."""
    if seq.shape[2] == 1:
        seq = seq.repeat(1, 1, 3, 1, 1)
    cp = _ring(seq[seq_len - 1], _CP_RGB, padding)
    seq[0] = _ring(seq[0], _START_RGB, padding)
    for i in range(seq_len - 1, output_len):
        seq[i] = cp
    return seq


def add_samples_cp_border(samples: torch.Tensor, seq_len: int, output_len: int, padding: int = 3):
    """Mark each sample row: orange ring on frame 0, red ring on the LAST
    generated frame. samples: (nsample,t,b,c,h,w)."""
    if samples.shape[3] == 1:
        samples = samples.repeat(1, 1, 1, 3, 1, 1)
    samples[:, 0] = _ring(samples[:, 0], _START_RGB, padding)
    samples[:, output_len - 1] = _ring(samples[:, output_len - 1], _CP_RGB, padding)
    return samples


class Skeleton3DVisualizer:
    """matplotlib-3D skeleton renderer -> per-frame RGB arrays
    (reference data/human36m/human36m.py:290-366)."""

    def __init__(self, parents, plot_3d_limit=(-2 * STD_SCALE, 2 * STD_SCALE),
                 show_joint=False, show_ticks=False, render=False):
        import matplotlib

        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        from mpl_toolkits.mplot3d import Axes3D  # noqa: F401

        self.parents = parents
        self.plot_3d_limit = list(plot_3d_limit) if plot_3d_limit is not None else None
        self.camera_azimuth = [70, 70, 110, 110]
        self.fig = plt.figure(figsize=(2, 2), dpi=64)
        self.fig.subplots_adjust(left=0, right=1, top=1, bottom=0, wspace=0, hspace=0)
        ax = self.fig.add_subplot(1, 1, 1, projection="3d")
        self.ax = ax
        if not show_ticks:
            ax.set_xticklabels([])
            ax.set_yticklabels([])
            ax.set_zticklabels([])
        if self.plot_3d_limit is not None:
            ax.set_xlim3d(*self.plot_3d_limit[::-1])
            ax.set_ylim3d(*self.plot_3d_limit)
            ax.set_zlim3d(*self.plot_3d_limit[::-1])
        self.line_3d = []
        for l_i in range(len(self.parents) - 1):
            if l_i in (0, 1, 2, 13, 14, 15):     # right limbs, 17-joint skeleton
                color = "r"
            elif l_i in (3, 4, 5, 10, 11, 12):   # left limbs
                color = "b"
            else:
                color = "g"
            self.line_3d.append(ax.plot([0, 1], [0, 1], [0, 1], zdir="z", c=color, linewidth=3))

    def _fig_to_rgb(self) -> np.ndarray:
        self.fig.canvas.draw()
        w, h = self.fig.canvas.get_width_height()
        buf = np.frombuffer(self.fig.canvas.buffer_rgba(), dtype=np.uint8).reshape(h, w, 4)
        img = buf[:, :, :3].copy()
        crop = 15
        return img[crop : h - crop, crop : w - crop]

    def set_data(self, pose_3d: np.ndarray, camera_view: int) -> np.ndarray:
        """pose_3d: (T, J, 3). Returns (T, H, W, 3) uint8."""
        self.ax.view_init(elev=15.0, azim=self.camera_azimuth[int(camera_view) % 4])
        imgs = []
        for seq_3d in pose_3d:
            for d_i, d_3d in enumerate(seq_3d):
                if d_i == 0:
                    continue
                p = self.parents[d_i]
                self.line_3d[d_i - 1][0].set_xdata([d_3d[0], seq_3d[p, 0]])
                self.line_3d[d_i - 1][0].set_ydata([d_3d[2], seq_3d[p, 2]])
                self.line_3d[d_i - 1][0].set_3d_properties([d_3d[1], seq_3d[p, 1]], zdir="z")
            imgs.append(self._fig_to_rgb())
        return np.array(imgs)


@torch.no_grad()
def vis_seq(model, x, epoch: int, output_len: int, model_mode: str = "full",
            recon_mode: Optional[str] = None, skip_frame: bool = True,
            h36m_visualizer=None, writer=None, opt=None):
    """Generate nsample sequences, paint borders, save PNG grid + GIF, log.

    Mirrors reference misc/visualize.py:90-272.
    """
    cfg = opt
    nsample = cfg.nsample
    grid_padding = 0
    nrow_per_block = 6

    gen_samples = []
    start_ix = 0

    if cfg.dataset == "h36m":
        n_block = min(cfg.batch_size, 5)
        pose_2d, pose_3d, camera_view = x
        pose_2d, pose_3d = pose_2d[:, :n_block], pose_3d[:, :n_block]
        camera_view = camera_view[:n_block]
        x = (pose_2d, pose_3d, camera_view)
        gt_seq = [pose_3d[i] for i in range(len(pose_3d))]
        seq_len = len(pose_3d)
        cp_ix = seq_len - 1
        x_cp = pose_3d[cp_ix]
    else:
        n_block = min(cfg.batch_size, 10)
        gt_seq = [x[i] for i in range(len(x))]
        seq_len = len(x)
        cp_ix = seq_len - 1
        x_cp = x[cp_ix]

    eval_cp_ix = output_len - 1
    for i in range(seq_len, output_len):
        gt_seq.append(x_cp)

    for _ in range(nsample):
        gen_seq = model.p2p_generate(
            x, output_len, eval_cp_ix, start_ix=start_ix, cp_ix=cp_ix,
            model_mode=model_mode, skip_frame=skip_frame,
        )
        gen_samples.append(torch.stack(gen_seq))

    if cfg.dataset == "h36m":
        # render skeleton frames to images
        for s in range(nsample):
            seq = gen_samples[s]
            imgs_inb = []
            for b in range(seq.shape[1]):
                imgs_inb.append(
                    h36m_visualizer.set_data(seq[:, b].cpu().numpy(), int(camera_view[b]))
                )
            imgs_inb = list(zip(*imgs_inb))
            frames = [
                torch.tensor(np.stack(f).astype(np.float32) / 255.0).permute(0, 3, 1, 2)
                for f in imgs_inb
            ]
            gen_samples[s] = torch.stack(frames)
        gt = torch.stack([g if torch.is_tensor(g) else torch.tensor(g) for g in gt_seq])
        imgs_inb = []
        for b in range(gt.shape[1]):
            imgs_inb.append(h36m_visualizer.set_data(gt[:, b].cpu().numpy(), int(camera_view[b])))
        imgs_inb = list(zip(*imgs_inb))
        gt_seq = [
            torch.tensor(np.stack(f).astype(np.float32) / 255.0).permute(0, 3, 1, 2)
            for f in imgs_inb
        ]

    gen_samples = torch.stack(gen_samples).cpu().float()
    r_len = max(seq_len, output_len)
    gt_seq = torch.stack([g.cpu().float() for g in gt_seq])

    gt_seq = add_gt_cp_border(gt_seq, seq_len=seq_len, output_len=output_len)
    gen_samples = add_samples_cp_border(gen_samples, seq_len=seq_len, output_len=output_len)

    img_canvas = []
    all_row_block = []
    for i in range(n_block):
        row_block = [gt_seq[:, i]]
        s_list = [1 % nsample] + list(np.random.randint(nsample, size=nrow_per_block - 2))
        for s in s_list:
            sample_j = gen_samples[s]
            if r_len > len(sample_j):
                pad_img = sample_j[eval_cp_ix].unsqueeze(0).repeat(
                    r_len - len(sample_j), 1, 1, 1, 1
                )
                sample_j = torch.cat([sample_j, pad_img])
            row_block.append(sample_j[:, i])
        row_block = torch.stack(row_block)
        rows = [make_grid(rb, nrow=r_len, padding=grid_padding) for rb in row_block]
        img_canvas.append(torch.cat(rows, dim=1))
        all_row_block.append(row_block)

    img = torch.cat(img_canvas, dim=1)
    os.makedirs(os.path.join(cfg.log_dir, "gen_vis"), exist_ok=True)
    if recon_mode in ("train", "test"):
        stem = f"recon_{recon_mode}-model_{model_mode}-len_{output_len}-epoch_{epoch}"
    else:
        stem = f"gen-model_{model_mode}-len-{output_len}-epoch_{epoch}"
    save_image(img, os.path.join(cfg.log_dir, "gen_vis", stem + ".png"))

    all_row_block = torch.stack(all_row_block)  # n_block, nrow_per_block, t, c, h, w
    gif_frames = []
    vid_tensor = []
    for t in range(r_len):
        cols = [
            make_grid(all_row_block[:, c, t], nrow=1, padding=grid_padding)
            for c in range(nrow_per_block)
        ]
        frame_t = torch.cat(cols, dim=2)
        vid_tensor.append(frame_t)
        gif_frames.append(to_uint8_hwc(frame_t))
    save_gif(os.path.join(cfg.log_dir, "gen_vis", stem + ".gif"), gif_frames)

    if writer is not None:
        if recon_mode in ("train", "test"):
            img_tag = f"{model_mode}/{recon_mode}-Gen"
            vid_tag = f"{model_mode}/{recon_mode}-Video"
        else:
            img_tag = f"{model_mode}/Gen{output_len}"
            vid_tag = f"{model_mode}/GenVideo{output_len}"
        writer.add_image(img_tag, img, epoch)
        writer.add_video(vid_tag, vid_tensor, epoch, fps=2)
    return img
