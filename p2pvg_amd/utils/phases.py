"""Per-phase timing harness.

The reference has no profiling beyond wall-clock around eval; this measures
one eager training iteration split into the phases that matter for the
MI355X schedule: HtoD, forward (the full unrolled recurrence + losses),
backward phase 1 (non-prior), non-prior Adam steps, backward phase 2
(prior, pruned), prior Adam step. CUDA timing uses events; CPU falls back
to perf_counter. The hipGraph path fuses all phases into one replay, so
this harness is eager-only by construction — use it to find which phase to
attack, rocprofv3 (profiles/) to see inside the kernels.

CLI: python -m p2pvg_amd.utils.phases --batch 64 [--backbone vgg ...]
"""
from __future__ import annotations

import time
from typing import Dict, List

import torch


class _Timer:
    def __init__(self, device: torch.device):
        self.cuda = device.type == "cuda"
        self.marks: List = []

    def mark(self, name: str):
        if self.cuda:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self.marks.append((name, ev))
        else:
            self.marks.append((name, time.perf_counter()))

    def deltas(self) -> Dict[str, float]:
        if self.cuda:
            torch.cuda.synchronize()
            out = {}
            for (n0, e0), (_, e1) in zip(self.marks, self.marks[1:]):
                out[n0] = e0.elapsed_time(e1)
            return out
        return {
            n0: (t1 - t0) * 1e3
            for (n0, t0), (_, t1) in zip(self.marks, self.marks[1:])
        }


def measure_phase_times(model, x_host, iters: int = 3, warmup: int = 2,
                        amp: bool = False) -> Dict[str, float]:
    """Mean per-phase milliseconds over `iters` eager steps.

    x_host: (T,B,...) batch on the HOST (pinned if CUDA) so the HtoD phase
    is real. Returns {phase: ms} with phases htod/forward/bwd_nonprior/
    step_nonprior/bwd_prior/step_prior.
    """
    import contextlib

    device = next(model.parameters()).device
    acc: Dict[str, float] = {}
    n_timed = 0
    for it in range(warmup + iters):
        t = _Timer(device)
        model.zero_grad(set_to_none=False)
        t.mark("htod")
        x = x_host.to(device, non_blocking=True)
        t.mark("forward")
        ctx = (
            torch.autocast("cuda", dtype=torch.bfloat16)
            if amp and device.type == "cuda"
            else contextlib.nullcontext()
        )
        with ctx:
            seq_len = len(x)
            plan = model.plan_step(seq_len)
            from ..models.p2p import gather_frames

            idx = torch.tensor(plan.proc, device=device)
            prev = gather_frames(x, idx - 1)
            cur = gather_frames(x, idx)
            tun = torch.as_tensor(plan.tun).to(device).view(-1, 1, 1)
            dts = torch.as_tensor(plan.dts).to(device).view(-1, 1, 1)
            mse, kld, cpc, align = model._compute_losses(prev, cur, tun, dts, plan)
        cfg = model.cfg
        loss1 = mse + kld * cfg.beta + align * cfg.weight_align
        loss2 = kld + cpc * cfg.weight_cpc
        nonprior, prior = model._param_groups()
        t.mark("bwd_nonprior")
        torch.autograd.backward(loss1, inputs=nonprior, retain_graph=True)
        t.mark("bwd_prior")
        torch.autograd.backward(loss2, inputs=prior)
        t.mark("step_nonprior")
        model.update_model_without_prior()
        t.mark("step_prior")
        model.update_prior()
        t.mark("end")
        if it >= warmup:
            for k, v in t.deltas().items():
                acc[k] = acc.get(k, 0.0) + v
            n_timed += 1
    return {k: v / max(n_timed, 1) for k, v in acc.items()}


def main():
    import argparse

    import numpy as np

    from ..core import Config
    from ..models import P2PModel

    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--seq_len", type=int, default=30)
    p.add_argument("--backbone", type=str, default="vgg")
    p.add_argument("--image_width", type=int, default=64)
    p.add_argument("--dataset", type=str, default="bair")
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--device", type=str, default=None)
    args = p.parse_args()

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    use_cuda = device.startswith("cuda")
    cfg = Config(
        dataset=args.dataset, backbone=args.backbone,
        image_width=args.image_width,
        channels=3 if args.dataset in ("bair", "weizmann") else 1,
        batch_size=args.batch, max_seq_len=args.seq_len, skip_prob=0.0,
        dtype="bf16" if use_cuda else "fp32", device=device,
    )
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(cfg).to(device)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    x = torch.rand(cfg.max_seq_len, cfg.batch_size, cfg.channels,
                   cfg.image_width, cfg.image_width)
    if use_cuda:
        t, b, c, h, w = x.shape
        x = x.permute(0, 1, 3, 4, 2).contiguous().permute(0, 1, 4, 2, 3)
        x = x.pin_memory()
    times = measure_phase_times(model, x, iters=args.iters, amp=use_cuda)
    total = sum(times.values())
    print(f"{'phase':<14} {'ms':>10} {'%':>6}")
    for k, v in times.items():
        print(f"{k:<14} {v:>10.2f} {100 * v / total:>6.1f}")
    print(f"{'total':<14} {total:>10.2f}")


if __name__ == "__main__":
    main()
