#!/usr/bin/env python
"""Training CLI — reference-compatible entrypoint (reference train.py).

Same flags, log-dir naming scheme, per-epoch checkpoints + latest copy, and
qualitative eval cadence as the reference (train.py:33-282), with the
MI355X-native additions: --device/--dtype/--ddp/--use_graphs, device-side loss
accumulation (one DtoH per log interval instead of per step), atomic latest
checkpoint, and optional torchrun launch (one rank per GPU over RCCL).
"""
from __future__ import annotations

import argparse
import os
import random
import time
from datetime import datetime

import numpy as np
import torch

from p2pvg_amd import data as data_utils
from p2pvg_amd.core import add_config_args, config_from_args
from p2pvg_amd.models import P2PModel
from p2pvg_amd.utils import ScalarWriter, get_logger, save_checkpoint, store_cmd
from p2pvg_amd.utils.vis import Skeleton3DVisualizer, vis_seq, STD_SCALE


def build_log_dir(cfg) -> str:
    if cfg.ckpt:
        from p2pvg_amd.utils import config_from_states, load_checkpoint

        states = load_checkpoint(cfg.ckpt)
        return config_from_states(states).log_dir
    log_suffix = {
        "dataset": cfg.dataset,
        "cpc": cfg.weight_cpc,
        "align": cfg.weight_align,
        "skip_prob": cfg.skip_prob,
        "batch_size": cfg.batch_size,
        "backbone": cfg.backbone,
        "beta": cfg.beta,
        "g_dim": cfg.g_dim,
        "z_dim": cfg.z_dim,
        "rnn_size": cfg.rnn_size,
    }
    log_name = "P2PModel" + "".join(f"-{k}_{v}" for k, v in log_suffix.items())
    log_dir = f"{cfg.log_dir}-{log_name}"
    if cfg.test:
        log_dir = "logs/test-%s-%s" % (
            os.path.basename(log_dir),
            datetime.now().strftime("%Y-%m-%d_%H-%M"),
        )
    return log_dir


def setup_distributed(cfg):
    """One process per GPU over RCCL (torch.distributed 'nccl' backend on ROCm)."""
    if "RANK" not in os.environ:
        return 0, 1
    import torch.distributed as dist

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        cfg.device = f"cuda:{local_rank}"
    cfg.ddp = world > 1
    return rank, world


def main():
    parser = argparse.ArgumentParser()
    add_config_args(parser)
    args = parser.parse_args()
    cfg = config_from_args(args)

    if cfg.deterministic:
        # CI/debug mode. The hand-written gfx950 kernels are
        # deterministic BY CONSTRUCTION since the round-2 reduction redesign:
        # every cross-block reduction (conv BN-stats, wgrad split-K, BN
        # backward, channel sums, fused MSE/KL) stores per-block partials and
        # combines them serially — no fp32 atomics anywhere on the hot path —
        # so determinism no longer means abandoning the native kernels.
        # use_deterministic_algorithms covers the remaining ATen/hipBLASLt
        # dispatches (warn_only: bf16 GEMMs have no deterministic flag).
        torch.use_deterministic_algorithms(True, warn_only=True)

    rank, world = setup_distributed(cfg)
    is_main = rank == 0

    cfg.log_dir = build_log_dir(cfg)
    if is_main:
        os.makedirs(os.path.join(cfg.log_dir, "gen_vis"), exist_ok=True)
        store_cmd(cfg.log_dir)
    writer = ScalarWriter(cfg.log_dir) if is_main else None
    logger = get_logger(os.path.join(cfg.log_dir, "logs")) if is_main else get_logger(
        os.devnull, displaying=False, saving=False
    )
    logger.info(str(cfg.to_dict()))

    # seeding (reference train.py:125-127) with per-rank offset for DDP shards
    seed = cfg.seed + rank
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)

    device = torch.device(cfg.resolved_device())

    train_data, test_data = data_utils.load_dataset(cfg)
    train_generator = data_utils.get_data_generator(train_data, train=True, opt=cfg, rank=rank)
    test_generator = data_utils.get_data_generator(
        test_data, train=False, dynamic_length=True, opt=cfg, rank=rank
    )

    visualizer = None
    if cfg.dataset == "h36m":
        visualizer = Skeleton3DVisualizer(
            train_data.skeleton.parents(),
            plot_3d_limit=[-2 * STD_SCALE, 2 * STD_SCALE],
            show_joint=False,
            show_ticks=False,
        )

    model = P2PModel(cfg).to(device)
    if device.type == "cuda" and cfg.dataset != "h36m":
        # NHWC params/activations: the gfx950 conv kernels are NHWC-native
        # (and MIOpen igemm too on the fallback path)
        model = model.to(memory_format=torch.channels_last)

    start_epoch = 0
    if cfg.ckpt:
        start_epoch = model.load(pth=cfg.ckpt)
        logger.info(f"[*] loaded {cfg.ckpt}; continuing at epoch {start_epoch}")

    ddp_hook = None
    if cfg.ddp and world > 1:
        from p2pvg_amd.parallel import DDPGradSync

        ddp_hook = DDPGradSync(model, bucket_mb=cfg.bucket_mb)
        ddp_hook.broadcast_parameters()

    amp_dtype = torch.bfloat16 if cfg.dtype == "bf16" else None
    qual_lengths = [10, 30]

    stepper = None
    if cfg.use_graphs and device.type == "cuda":
        # With DDP the two RCCL all-reduce flush points are recorded INSIDE
        # the captured step (grad_sync runs in _backward_and_step). RCCL
        # supports stream-capture of collectives; if capture fails on any
        # rank the stepper falls back to the eager step for that shape key,
        # so --use_graphs is safe to combine with --ddp (measured eager vs
        # graphed numbers: docs/ROADMAP.md).
        from p2pvg_amd.runtime import GraphedTrainStep

        if world > 1:
            logger.info("[*] hipGraph capture with RCCL collectives recorded "
                        "in-graph (falls back to eager per shape key on "
                        "capture failure)")
        stepper = GraphedTrainStep(model, amp_dtype=amp_dtype)

    for epoch in range(start_epoch, cfg.nepochs):
        model.train()
        # device-side accumulators: one DtoH per log interval
        acc = torch.zeros(4, device=device)
        t0 = time.time()
        frames = 0

        for i in range(cfg.epoch_size):
            x = next(train_generator)
            if stepper is not None:
                losses = stepper.step(x)
            else:
                model.zero_grad(set_to_none=False)
                if amp_dtype is not None and device.type == "cuda":
                    with torch.autocast("cuda", dtype=amp_dtype):
                        losses = model(x, 0, len(x) - 1)
                else:
                    losses = model(x, 0, len(x) - 1)
            acc += torch.stack(list(losses))
            frames += (len(x) if not isinstance(x, tuple) else len(x[1])) * cfg.batch_size

            if world > 1 and i % 50 == 0 and i != 0:
                # average the 4 loss scalars across ranks for logging
                # (collective: must run on every rank) — SURVEY §5.8
                import torch.distributed as dist

                lg = acc.clone()
                dist.all_reduce(lg)
                lg /= world
            else:
                lg = acc
            if is_main and i % 50 == 0 and i != 0:
                step = epoch * cfg.epoch_size + i
                vals = (lg / (i + 1)).cpu()
                for tag, v in zip(("mse", "kld", "cpc", "align"), vals):
                    writer.add_scalar(f"Train/{tag}", v, step)
                if cfg.log_histograms:
                    for name, param in model.named_parameters():
                        if param.requires_grad:
                            writer.add_histogram_summary(name.replace(".", "/"), param, step)

        vals = (acc / cfg.epoch_size).cpu()
        dt = time.time() - t0
        logger.info(
            "[%02d] mse: %.5f | kld: %.5f | align: %.5f | cpc: %.5f | %.1f frames/s"
            % (epoch, vals[0], vals[1], vals[3], vals[2], frames / dt)
        )

        # quantitative eval: end-frame SSIM (the BASELINE quality metric the
        # reference never implemented, misc/metrics.py stub)
        if is_main and cfg.dataset != "h36m" and (epoch + 1) % cfg.quan_iter == 0:
            from p2pvg_amd.utils import end_frame_ssim

            model.eval()
            with torch.no_grad():
                x = next(test_generator)
                gen = model.p2p_generate(x, len(x), len(x) - 1, model_mode="full",
                                         skip_frame=False)
                s = end_frame_ssim(gen, x[len(x) - 1]).mean()
                writer.add_scalar("Eval/end_frame_ssim", s, epoch)
                logger.info("[%02d] end-frame SSIM (full mode): %.4f" % (epoch, s))
            model.train()

        # qualitative eval (reference train.py:246-272)
        if is_main and (epoch + 1) % cfg.qual_iter == 0:
            model.eval()
            with torch.no_grad():
                x = next(test_generator)
                length_to_gen = x[1].shape[0] if cfg.dataset == "h36m" else len(x)
                for mode in ("full", "posterior", "prior"):
                    vis_seq(model, x, epoch, length_to_gen, model_mode=mode,
                            recon_mode="test", skip_frame=False,
                            h36m_visualizer=visualizer, writer=writer, opt=cfg)
                for length_to_gen in qual_lengths:
                    for mode in ("full", "posterior", "prior"):
                        vis_seq(model, x, epoch, length_to_gen, model_mode=mode,
                                skip_frame=False, h36m_visualizer=visualizer,
                                writer=writer, opt=cfg)
            model.train()

        if is_main:
            fname = os.path.join(cfg.log_dir, f"model_{epoch}.pth")
            save_checkpoint(model, epoch, fname,
                            latest_name=os.path.join(cfg.log_dir, "model.pth"))
            logger.info(f"[*] model saved at: {fname}")


if __name__ == "__main__":
    main()
