#!/usr/bin/env python
"""Generation CLI — reference-compatible entrypoint (reference generate.py).

Loads a checkpoint (ours or a reference-written .pth), rebuilds the model from
the checkpoint's own config, reads an input video (.mp4/.gif via PIL, or a
directory of frames, or --start_img/--end_img image pair — the reference
declared that path but never defined the flags, generate.py:93-96), and
generates lengths [10, 20, 30] x 5 samples, saving PNG rows and GIFs.
"""
from __future__ import annotations

import argparse
import os
import random

import numpy as np
import torch

from p2pvg_amd.models import P2PModel
from p2pvg_amd.utils import config_from_states, load_checkpoint
from p2pvg_amd.utils.image import make_grid, save_gif, save_image, to_uint8_hwc
from p2pvg_amd.utils.vis import add_gt_cp_border, add_samples_cp_border


def _read_video_mp4(vid_name: str) -> list:
    """Decode an .mp4/.avi/.mov (reference generate.py:29-39 reads .mp4 via
    imageio/ffmpeg). Tries, in order: imageio(.v3), PyAV, an ffmpeg binary
    (rawvideo pipe). All are optional dependencies; a clear error names them
    when none is present."""
    frames = []
    try:  # imageio v3 or v2 (bundles imageio-ffmpeg when installed)
        try:
            import imageio.v3 as iio

            for fr in iio.imiter(vid_name, plugin="pyav"):
                frames.append(np.asarray(fr, dtype=np.float32) / 255.0)
        except Exception:
            import imageio

            for fr in imageio.get_reader(vid_name):
                frames.append(np.asarray(fr, dtype=np.float32) / 255.0)
        if frames:
            return frames
    except ImportError:
        pass
    try:  # PyAV directly
        import av

        with av.open(vid_name) as container:
            for fr in container.decode(video=0):
                frames.append(
                    np.asarray(fr.to_rgb().to_ndarray(), dtype=np.float32) / 255.0
                )
        if frames:
            return frames
    except ImportError:
        pass
    import shutil
    import subprocess

    ffmpeg = shutil.which("ffmpeg")
    ffprobe = shutil.which("ffprobe")
    if ffmpeg and ffprobe:
        probe = subprocess.run(
            [ffprobe, "-v", "error", "-select_streams", "v:0", "-show_entries",
             "stream=width,height", "-of", "csv=p=0", vid_name],
            capture_output=True, text=True, check=True,
        )
        w, h = (int(v) for v in probe.stdout.strip().split(",")[:2])
        raw = subprocess.run(
            [ffmpeg, "-v", "error", "-i", vid_name, "-f", "rawvideo",
             "-pix_fmt", "rgb24", "-"],
            capture_output=True, check=True,
        ).stdout
        arr = np.frombuffer(raw, dtype=np.uint8).reshape(-1, h, w, 3)
        return [f.astype(np.float32) / 255.0 for f in arr]
    raise RuntimeError(
        f"cannot decode {vid_name!r}: no mp4 backend available. Install one of "
        "imageio[ffmpeg], av (PyAV), or an ffmpeg binary on PATH — or pass a "
        ".gif / frame directory instead."
    )


def read_video(vid_name: str) -> torch.Tensor:
    """Read a video (.mp4/.gif), or a frame directory -> (t, 1, c, h, w) in [0,1]."""
    frames = []
    if os.path.isdir(vid_name):
        from PIL import Image

        for f in sorted(os.listdir(vid_name)):
            if f.lower().endswith((".png", ".jpg", ".jpeg")):
                with Image.open(os.path.join(vid_name, f)) as im:
                    frames.append(np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0)
    elif vid_name.lower().endswith((".mp4", ".avi", ".mov", ".mkv", ".webm")):
        frames = _read_video_mp4(vid_name)
    else:
        from PIL import Image, ImageSequence

        with Image.open(vid_name) as im:
            for frame in ImageSequence.Iterator(im):
                frames.append(np.asarray(frame.convert("RGB"), dtype=np.float32) / 255.0)
    if not frames:
        raise ValueError(f"no frames read from {vid_name}")
    t = torch.from_numpy(np.stack(frames)).permute(0, 3, 1, 2)
    return t.unsqueeze(1)


def read_image_pair(start_img: str, end_img: str) -> torch.Tensor:
    from PIL import Image

    out = []
    for p in (start_img, end_img):
        with Image.open(p) as im:
            out.append(
                torch.from_numpy(
                    np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0
                ).permute(2, 0, 1)
            )
    return torch.stack(out).unsqueeze(1)


def multi_cp_generate(model, control_frames, seg_len: int):
    """Chain point-to-point segments through a list of control frames,
    carrying hidden state across segments (the reference README's
    'multiple control points generation' capability, realized through
    p2p_generate's init_hidden flag)."""
    out = []
    init = True
    for a, b in zip(control_frames[:-1], control_frames[1:]):
        seq = torch.stack([a, b])  # (2, B, C, H, W)
        gen = model.p2p_generate(seq, seg_len, seg_len - 1, model_mode="full",
                                 init_hidden=init)
        init = False
        out.extend(gen if not out else gen[1:])
    return out


def loop_generate(model, start_frame, end_frame, seg_len: int):
    """A -> B -> A loop (the reference README's 'loop generation')."""
    return multi_cp_generate(model, [start_frame, end_frame, start_frame], seg_len)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--ckpt", type=str, required=True, help="model .pth file")
    parser.add_argument("--video", type=str, default="", help=".gif video or frame dir")
    parser.add_argument("--start_img", type=str, default="")
    parser.add_argument("--end_img", type=str, default="")
    parser.add_argument("--output_root", type=str, default="gen_outputs")
    parser.add_argument("--seed", type=int, default=1)
    parser.add_argument("--device", type=str, default="auto")
    parser.add_argument("--multi_cp", action="store_true",
                        help="chain segments through every input frame as a control point")
    parser.add_argument("--loop", action="store_true",
                        help="loop generation: first -> last -> first")
    parser.add_argument("--segment_len", type=int, default=10,
                        help="frames per segment for --multi_cp / --loop")
    args = parser.parse_args()

    states = load_checkpoint(args.ckpt)
    cfg = config_from_states(states)
    cfg.batch_size = 1
    cfg.device = args.device

    random.seed(args.seed)
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(args.seed)

    device = torch.device(cfg.resolved_device())
    model = P2PModel(cfg).to(device)
    model.load(states=states)
    model.eval()

    nsamples, ndisplays = 5, 5
    gen_lengths = [10, 20, 30]

    if args.video:
        seq = read_video(args.video)
    elif args.start_img:
        assert args.end_img, "--start_img requires --end_img"
        seq = read_image_pair(args.start_img, args.end_img)
    else:
        raise SystemExit("provide --video or --start_img/--end_img")

    if cfg.channels == 1 and seq.shape[2] == 3:
        seq = seq.mean(dim=2, keepdim=True)
    if seq.shape[-2:] != (cfg.image_width, cfg.image_width):
        # arbitrary input resolutions: resize to the model's training size
        t_, b_, c_ = seq.shape[:3]
        seq = torch.nn.functional.interpolate(
            seq.reshape(t_ * b_, c_, *seq.shape[-2:]),
            size=(cfg.image_width, cfg.image_width),
            mode="bilinear", align_corners=False,
        ).reshape(t_, b_, c_, cfg.image_width, cfg.image_width)
    seq = seq.to(device)
    seq_len = len(seq)

    os.makedirs(args.output_root, exist_ok=True)

    if args.multi_cp or args.loop:
        with torch.no_grad():
            if args.loop:
                gen = loop_generate(model, seq[0], seq[seq_len - 1], args.segment_len)
                stem = "loopgen"
            else:
                gen = multi_cp_generate(model, list(seq), args.segment_len)
                stem = "mulcpgen"
            gen = torch.stack(gen).cpu().float()
            save_image(
                make_grid(gen[:, 0], nrow=len(gen), padding=0),
                os.path.join(args.output_root, f"{stem}.png"),
            )
            save_gif(
                os.path.join(args.output_root, f"{stem}.gif"),
                [to_uint8_hwc(gen[t, 0]) for t in range(len(gen))],
            )
            print(f"[*] wrote {stem} outputs ({len(gen)} frames) to {args.output_root}")
        return

    with torch.no_grad():
        for length_to_gen in gen_lengths:
            output_cp_ix = length_to_gen - 1
            samples = []
            for _ in range(nsamples):
                out = model.p2p_generate(seq, length_to_gen, output_cp_ix, model_mode="full")
                samples.append(torch.stack(out))
            samples = torch.stack(samples)

            idx = np.random.choice(len(samples), ndisplays, replace=False)
            samples_to_save = samples[idx].cpu().float()

            padded_seq = seq.clone().cpu().float()
            x_cp = padded_seq[seq_len - 1]
            if length_to_gen > seq_len:
                pad = x_cp.unsqueeze(0).repeat(length_to_gen - seq_len, 1, 1, 1, 1)
                padded_seq = torch.cat([padded_seq, pad], dim=0)

            seq_b = add_gt_cp_border(padded_seq, seq_len, length_to_gen)
            samples_to_save = add_samples_cp_border(samples_to_save, seq_len, length_to_gen)

            save_image(
                make_grid(seq_b[:, 0], nrow=len(seq_b), padding=0),
                os.path.join(args.output_root, f"len_{length_to_gen}-gt.png"),
            )

            block = []
            for ix, s in enumerate(samples_to_save):
                row = make_grid(s[:, 0], nrow=len(s), padding=0)
                save_image(row, os.path.join(args.output_root, f"len_{length_to_gen}-gen_{ix:03d}.png"))
                block.append(row)
            save_image(
                torch.cat(block, 1),
                os.path.join(args.output_root, f"len_{length_to_gen}-gen_full.png"),
            )

            for ix, s in enumerate(samples_to_save):
                frames = [to_uint8_hwc(s[t, 0]) for t in range(len(s))]
                save_gif(os.path.join(args.output_root, f"len_{length_to_gen}-gen_{ix:03d}.gif"), frames)

            gifs = [
                to_uint8_hwc(make_grid(samples_to_save[:, t, 0], nrow=ndisplays, padding=0))
                for t in range(length_to_gen)
            ]
            save_gif(os.path.join(args.output_root, f"len_{length_to_gen}-gen_full.gif"), gifs)
            print(f"[*] wrote length-{length_to_gen} outputs to {args.output_root}")


if __name__ == "__main__":
    main()
