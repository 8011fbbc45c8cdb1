"""P2PModel: conditional VAE over time for point-to-point video generation.

Capability parity with reference models/p2p_model.py (training step semantics
documented in SURVEY §2.2): three LSTM stacks (posterior / prior / frame
predictor) over a backbone encoder/decoder with U-Net skips, a global
descriptor of the control-point frame, two scalar time signals per step,
skip-frame training, and four losses (MSE recon, KL(q||p), CPC at the control
point, latent alignment) with a two-phase update (prior is updated on
kld + w_cpc*cpc only; everything else on mse + beta*kld + w_align*align).

MI355X-native differences:
- Device-agnostic (no ctor-time .cuda(); contrast reference models/lstm.py:58).
- The two-phase backward uses gradient-routed pruned traversals
  (`torch.autograd.backward(..., inputs=...)`): phase 2 walks only the
  kld->prior and cpc->prior paths instead of re-walking the whole unrolled
  graph as the reference does (reference models/p2p_model.py:262,268).
  `backward_mode="reference"` reproduces the literal double full backward.
- Losses are accumulated on device; no per-iteration DtoH sync (the reference
  syncs every step, models/p2p_model.py:271).
- bf16 autocast compute with fp32 BatchNorm stats and fp32 loss accumulation
  when cfg.dtype == "bf16".
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.nn as nn

from .. import ops
from .backbones import build_backbone
from .lstm import gaussian_lstm, lstm


from dataclasses import dataclass


@dataclass(frozen=True)
class StepPlan:
    """Host-side plan of one training step (which steps run, time signals)."""

    seq_len: int
    cp_ix: int
    proc: tuple          # processed step indices i (i=1..cp_ix, minus skips)
    tun: np.ndarray      # time_until_cp per processed step
    dts: np.ndarray      # delta_time per processed step
    unpack: tuple        # per processed step: does this step refresh `skip`?

    @property
    def graph_key(self):
        return (self.seq_len, len(self.proc), self.unpack)


def gather_frames(x: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """index_select frames along time, preserving per-frame channels_last."""
    if x.dim() == 5 and x[0].is_contiguous(memory_format=torch.channels_last):
        xn = x.permute(0, 1, 3, 4, 2)          # (T,B,H,W,C) view of NHWC storage
        sel = xn.index_select(0, idx)          # contiguous (n,B,H,W,C)
        return sel.permute(0, 1, 4, 2, 3)      # (n,B,C,H,W), channels_last frames
    return x.index_select(0, idx)


def init_weights(m: nn.Module) -> None:
    """N(0, 0.02) conv/linear, N(1, 0.02) batchnorm (reference misc/utils.py:157-164)."""
    classname = m.__class__.__name__
    if classname.find("Conv") != -1 or classname.find("Linear") != -1:
        m.weight.data.normal_(0.0, 0.02)
        if getattr(m, "bias", None) is not None:
            m.bias.data.fill_(0)
    elif classname.find("BatchNorm") != -1:
        m.weight.data.normal_(1.0, 0.02)
        m.bias.data.fill_(0)


class P2PModel(nn.Module):
    NONPRIOR = ("frame_predictor", "posterior", "encoder", "decoder")

    def __init__(self, cfg):
        super().__init__()
        self.cfg = cfg
        g, z = cfg.g_dim, cfg.z_dim

        # the two scalar time signals (+1+1) ride along every LSTM input
        self.frame_predictor = lstm(
            g + z + 2, g, cfg.rnn_size, cfg.predictor_rnn_layers, cfg.batch_size
        )
        self.posterior = gaussian_lstm(
            g + g + 2, z, cfg.rnn_size, cfg.posterior_rnn_layers, cfg.batch_size
        )
        self.prior = gaussian_lstm(
            g + g + 2, z, cfg.rnn_size, cfg.prior_rnn_layers, cfg.batch_size
        )
        self.encoder, self.decoder = build_backbone(cfg)

        self.mse_criterion = nn.MSELoss()
        self.align_criterion = nn.MSELoss()

        self.apply_init_weights()
        self.init_optimizer()

    # -- setup ------------------------------------------------------------

    def apply_init_weights(self):
        for name in ("frame_predictor", "posterior", "prior", "encoder", "decoder"):
            getattr(self, name).apply(init_weights)

    def init_optimizer(self):
        cfg = self.cfg
        from ..optim import make_adam

        device = cfg.resolved_device() if hasattr(cfg, "resolved_device") else "cpu"
        for name in ("frame_predictor", "posterior", "prior", "encoder", "decoder"):
            opt = make_adam(
                getattr(self, name).parameters(), lr=cfg.lr,
                betas=(cfg.beta1, 0.999), device=device,
                capturable=bool(getattr(cfg, "use_graphs", False)),
            )
            setattr(self, f"{name}_optimizer", opt)

    def init_hidden(self, batch_size: int = 1, device=None, dtype=None):
        self.frame_predictor.init_hidden(batch_size, device, dtype)
        self.posterior.init_hidden(batch_size, device, dtype)
        self.prior.init_hidden(batch_size, device, dtype)

    # -- pieces ------------------------------------------------------------

    def kl_criterion(self, mu1, logvar1, mu2, logvar2):
        return ops.gaussian_kl(mu1, logvar1, mu2, logvar2, float(self.cfg.batch_size))

    def get_global_descriptor(self, x, start_ix: int = 0, cp_ix: Optional[int] = None):
        """Encode the control-point (end) frame into the global descriptor."""
        if cp_ix is None:
            cp_ix = len(x) - 1
        x_cp = x[cp_ix]
        h_cp = self.encoder(x_cp)[0]
        return x_cp, h_cp

    def plan_step(self, seq_len: int) -> "StepPlan":
        """Host-side step plan: draw the skip gate and derive the processed
        step indices and the two time-signal values per processed step.

        Skip-gate semantics match reference models/p2p_model.py:209-222: one
        pre-drawn U(0,1) per candidate step; never skips i==1 or the
        control-point step; at most seq_len*skip_prob skips; skipped steps
        freeze hidden state and widen the next delta_time.
        """
        cfg = self.cfg
        cp_ix = seq_len - 1
        probs = np.random.uniform(0, 1, seq_len - 1)
        proc, tun, dts = [], [], []
        prev_i = 0
        skip_count = 0
        max_skip = seq_len * cfg.skip_prob
        for i in range(1, seq_len):
            if (
                probs[i - 1] <= cfg.skip_prob
                and i >= cfg.n_past
                and skip_count < max_skip
                and i != 1
                and i != cp_ix
            ):
                skip_count += 1
                continue
            proc.append(i)
            tun.append((cp_ix - i + 1) / cp_ix)
            dts.append((i - prev_i) / cp_ix)
            prev_i = i
        unpack = tuple(
            bool(cfg.last_frame_skip or i <= cfg.n_past) for i in proc
        )
        return StepPlan(
            seq_len=seq_len,
            cp_ix=cp_ix,
            proc=tuple(proc),
            tun=np.asarray(tun, dtype=np.float32),
            dts=np.asarray(dts, dtype=np.float32),
            unpack=unpack,
        )

    def _compute_losses(self, prev_frames, cur_frames, tun_t, dt_t, plan):
        """Device-side training losses over the processed steps.

        prev_frames/cur_frames: (n, B, ...) gathered frames (prev_frames[k] =
        x[proc[k]-1], cur_frames[k] = x[proc[k]]); tun_t/dt_t: (n,1,1) device
        scalars. The structure depends only on (n, plan.unpack) — it is
        hipGraph-capturable with the frames/scalars as graph inputs.
        """
        cfg = self.cfg
        if prev_frames.is_cuda:
            # one cast+flip pass per step instead of per conv use; recorded
            # inside the hipGraph so replays refresh too
            from ..ops.conv import refresh_conv_shadows

            refresh_conv_shadows(self)
        n = len(plan.proc)
        batch_size = prev_frames[0].shape[0]
        device = prev_frames.device

        self.init_hidden(batch_size=batch_size, device=device)

        x_cp = cur_frames[n - 1]
        global_z = self.encoder(x_cp)[0]

        mse_loss = torch.zeros((), device=device)
        kld_loss = torch.zeros((), device=device)
        cpc_loss = torch.zeros((), device=device)
        align_loss = torch.zeros((), device=device)

        h_prev = None
        h_pred = None
        skip = None

        for k in range(n):
            if k > 0 and cfg.align_mode == "reference":
                # exact as-written semantics: batch row 0 of the PREVIOUS
                # iteration's latent broadcast against h_pred
                # (reference models/p2p_model.py:225 quirk)
                align_loss = align_loss + self.align_criterion(
                    h_prev[0].expand_as(h_pred), h_pred
                )

            time_until_cp = tun_t[k].expand(batch_size, 1)
            delta_time = dt_t[k].expand(batch_size, 1)

            h = self.encoder(prev_frames[k])
            h_target = self.encoder(cur_frames[k])[0]

            if plan.unpack[k]:
                h, skip = h
            else:
                h = h[0]

            if k > 0 and cfg.align_mode == "paper":
                # paper-intent: align the encoder latent of frame i-1 with
                # the predictor output that predicted that frame's latent
                align_loss = align_loss + self.align_criterion(h, h_pred)

            # LSTM inputs go in as (h, global, t, dt) TUPLES: the fused
            # embed kernel gathers the four parts directly (no concat
            # kernels); the CPU/torch path cats inside the stack
            h_cpaw = (h, global_z, time_until_cp, delta_time)
            h_target_cpaw = (h_target, global_z, time_until_cp, delta_time)

            zt, mu, logvar = self.posterior(h_target_cpaw)
            zt_p, mu_p, logvar_p = self.prior(h_cpaw)

            h_pred = self.frame_predictor(
                (h, zt, time_until_cp, delta_time)
            )
            x_pred = self.decoder([h_pred, skip])

            if k == n - 1:
                # control-point consistency (reference models/p2p_model.py:251-254)
                h_pred_p = self.frame_predictor(
                    (h, zt_p, time_until_cp, delta_time)
                )
                x_pred_p = self.decoder([h_pred_p, skip])
                cpc_loss = ops.frame_mse(x_pred_p, x_cp)

            mse_loss = mse_loss + ops.frame_mse(x_pred, cur_frames[k])
            kld_loss = kld_loss + self.kl_criterion(mu, logvar, mu_p, logvar_p)

            h_prev = h

        return mse_loss, kld_loss, cpc_loss, align_loss

    # -- training step ------------------------------------------------------

    def forward(self, x, start_ix: int = 0, cp_ix: int = -1):
        """One full training step: forward over the sequence, both backward
        phases, and all five optimizer steps. Returns the four per-step-mean
        losses as detached device scalars (no DtoH sync here)."""
        if isinstance(x, tuple):  # h36m yields (pose_2d, pose_3d, camera_view)
            x = x[1]

        device = x[0].device
        seq_len = len(x)
        plan = self.plan_step(seq_len)

        idx_cur = torch.tensor(plan.proc, device=device)
        prev_frames = gather_frames(x, idx_cur - 1)
        cur_frames = gather_frames(x, idx_cur)
        tun_t = torch.as_tensor(plan.tun).to(device).view(-1, 1, 1)
        dt_t = torch.as_tensor(plan.dts).to(device).view(-1, 1, 1)

        mse_loss, kld_loss, cpc_loss, align_loss = self._compute_losses(
            prev_frames, cur_frames, tun_t, dt_t, plan
        )
        self._backward_and_step(mse_loss, kld_loss, cpc_loss, align_loss)

        inv = 1.0 / seq_len
        return (
            (mse_loss * inv).detach(),
            (kld_loss * inv).detach(),
            (cpc_loss * inv).detach(),
            (align_loss * inv).detach(),
        )


    def _param_groups(self):
        nonprior = [p for n in self.NONPRIOR for p in getattr(self, n).parameters()]
        prior = list(self.prior.parameters())
        return nonprior, prior

    def zero_grads(self):
        """Zero every param grad in ONE fused dispatch group.

        Replaces nn.Module.zero_grad(set_to_none=False)'s per-tensor fills
        (~200 tiny kernels per step). Grads are materialized once with the
        params' own memory format and then kept at stable addresses — the
        custom conv/BN backward kernels ACCUMULATE into these buffers in
        kernel (ops/conv.py::_acc_target), so autograd's per-use accumulation
        adds never dispatch, and hipGraph capture sees fixed pointers."""
        gs = getattr(self, "_grad_list", None)
        if gs is None:
            for p in self.parameters():
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
            self._grad_list = gs = [p.grad for p in self.parameters()]
        torch._foreach_zero_(gs)

    def zero_grad(self, set_to_none: bool = True):
        # keep the stable-buffer discipline whatever the caller passes
        self.zero_grads()

    def _backward_and_step(self, mse_loss, kld_loss, cpc_loss, align_loss):
        cfg = self.cfg
        mode = getattr(cfg, "backward_mode", "pruned")
        loss1 = mse_loss + kld_loss * cfg.beta + align_loss * cfg.weight_align
        loss2 = kld_loss + cpc_loss * cfg.weight_cpc

        grad_sync = getattr(self, "grad_sync", None)
        nonprior, prior = self._param_groups()

        # NOTE on ordering: the reference steps the non-prior optimizers
        # BETWEEN the two backwards (reference models/p2p_model.py:259-269),
        # which under PyTorch-1.0 optimizers mutated the very weight storages
        # the retained graph had saved — so its phase-2 dgrad ran on
        # post-step weights. Modern autograd's version counter forbids that
        # (by design); both formulations here run both backwards on the
        # pre-step graph, then step. The effective update rule is unchanged:
        # non-prior params move by -lr*Adam(dL1), prior by -lr*Adam(dL2).

        from ..ops.conv import weight_grad_scope

        if mode == "reference":
            # two full-graph traversals, exactly as the reference pays them
            loss1.backward(retain_graph=True)
            stash = [p.grad.clone() if p.grad is not None else None for p in nonprior]
            for p in self.prior.parameters():
                if p.grad is not None:
                    p.grad.zero_()
            loss2.backward()
            # the reference stepped non-prior before loss2.backward, so the
            # L2 grads that leak into non-prior params never affect updates;
            # restore the phase-1 grads to reproduce that. copy_ (not rebind)
            # keeps grad buffer addresses stable for the fused zero/Adam.
            for p, g in zip(nonprior, stash):
                if g is not None:
                    if p.grad is None:
                        p.grad = g
                    else:
                        p.grad.copy_(g)
        else:
            # phase 1: grads of loss1 into everything but the prior. The
            # pruned traversal reaches the prior's backward nodes on the
            # h->encoder path, so the prior's in-kernel weight accumulation
            # must be scoped OFF here (the reference zeroes those grads
            # before phase 2, models/p2p_model.py:266).
            with weight_grad_scope(prior):
                torch.autograd.backward(loss1, inputs=nonprior,
                                        retain_graph=True)
            # phase 2: grads of loss2 into the prior only — the traversal
            # still reaches decoder/predictor nodes on the cpc path (their
            # dx is needed); their weight work is scoped off (discarded by
            # the update rule anyway).
            with weight_grad_scope(nonprior):
                torch.autograd.backward(loss2, inputs=prior)

        if grad_sync is not None:
            grad_sync.sync_nonprior()
        self.update_model_without_prior()
        if grad_sync is not None:
            grad_sync.sync_prior()
        self.update_prior()

    def update_prior(self):
        self.prior_optimizer.step()

    def update_model_without_prior(self):
        self.frame_predictor_optimizer.step()
        self.posterior_optimizer.step()
        self.encoder_optimizer.step()
        self.decoder_optimizer.step()

    def update_model(self):
        self.update_model_without_prior()
        self.update_prior()

    # -- generation ---------------------------------------------------------

    @torch.no_grad()
    def p2p_generate(
        self,
        x,
        len_output: int,
        eval_cp_ix: int,
        start_ix: int = 0,
        cp_ix: int = -1,
        model_mode: str = "full",
        skip_frame: bool = False,
        init_hidden: bool = True,
    ) -> List[torch.Tensor]:
        """Point-to-point generation (reference models/p2p_model.py:80-183).

        model_mode: full (posterior warm-up then prior), posterior, prior.
        Output length is decoupled from input length via the time signals.
        """
        cfg = self.cfg

        if isinstance(x, tuple):  # h36m
            x = x[1]

        batch_size = x[0].shape[0]
        device = x[0].device
        if x[0].is_cuda:
            from ..ops.conv import refresh_conv_shadows

            refresh_conv_shadows(self)

        gen_seq = [x[0]]
        x_in = x[0]

        if init_hidden:
            self.init_hidden(batch_size=batch_size, device=device)

        seq_len = len(x)
        cp_ix = seq_len - 1
        x_cp, global_z = self.get_global_descriptor(x, cp_ix=cp_ix)

        skip_prob = cfg.skip_prob
        prev_i = 0
        max_skip_count = seq_len * skip_prob
        skip_count = 0
        probs = np.random.uniform(0, 1, len_output - 1)
        skip = None

        for i in range(1, len_output):
            if (
                probs[i - 1] <= skip_prob
                and i >= cfg.n_past
                and skip_count < max_skip_count
                and i != 1
                and i != (len_output - 1)
                and skip_frame
            ):
                skip_count += 1
                gen_seq.append(torch.zeros_like(x_in))
                continue

            time_until_cp = torch.full(
                (batch_size, 1), (eval_cp_ix - i + 1) / eval_cp_ix,
                device=device, dtype=x_cp.dtype,
            )
            delta_time = torch.full(
                (batch_size, 1), (i - prev_i) / eval_cp_ix,
                device=device, dtype=x_cp.dtype,
            )
            prev_i = i

            h = self.encoder(x_in)
            if cfg.last_frame_skip or i == 1 or i < cfg.n_past:
                h, skip = h
            else:
                h = h[0]

            h_cpaw = (h, global_z, time_until_cp, delta_time)

            if i < cfg.n_past:
                # warm-up: drive the recurrence with ground truth
                h_target = self.encoder(x[i])[0]
                h_target_cpaw = (h_target, global_z, time_until_cp, delta_time)
                zt, _, _ = self.posterior(h_target_cpaw)
                zt_p, _, _ = self.prior(h_cpaw)
                if model_mode in ("posterior", "full"):
                    self.frame_predictor((h, zt, time_until_cp, delta_time))
                else:
                    self.frame_predictor((h, zt_p, time_until_cp, delta_time))
                x_in = x[i]
                gen_seq.append(x_in)
            else:
                if i < len(x):
                    h_target = self.encoder(x[i])[0]
                    h_target_cpaw = (h_target, global_z, time_until_cp,
                                     delta_time)
                else:
                    h_target_cpaw = h_cpaw

                zt, _, _ = self.posterior(h_target_cpaw)
                zt_p, _, _ = self.prior(h_cpaw)

                if model_mode == "posterior":
                    h = self.frame_predictor((h, zt, time_until_cp, delta_time))
                else:  # prior | full
                    h = self.frame_predictor((h, zt_p, time_until_cp, delta_time))

                x_in = self.decoder([h, skip])
                gen_seq.append(x_in)
        return gen_seq

    # -- checkpoint contract ----------------------------------

    def state_for_checkpoint(self, epoch: int) -> Dict:
        cfg_dict = self.cfg.to_dict()
        return {
            "encoder": self.encoder.state_dict(),
            "decoder": self.decoder.state_dict(),
            "frame_predictor": self.frame_predictor.state_dict(),
            "posterior": self.posterior.state_dict(),
            "prior": self.prior.state_dict(),
            "encoder_opt": self.encoder_optimizer.state_dict(),
            "decoder_opt": self.decoder_optimizer.state_dict(),
            "frame_predictor_opt": self.frame_predictor_optimizer.state_dict(),
            "posterior_opt": self.posterior_optimizer.state_dict(),
            "prior_opt": self.prior_optimizer.state_dict(),
            "epoch": epoch,
            "opt": cfg_dict,
        }

    def save(self, fname: str, epoch: int) -> None:
        torch.save(self.state_for_checkpoint(epoch), fname)

    def load(self, pth: Optional[str] = None, states: Optional[Dict] = None) -> int:
        if states is None:
            states = torch.load(pth, map_location="cpu", weights_only=False)
        self.encoder.load_state_dict(states["encoder"])
        self.decoder.load_state_dict(states["decoder"])
        self.frame_predictor.load_state_dict(states["frame_predictor"])
        self.posterior.load_state_dict(states["posterior"])
        self.prior.load_state_dict(states["prior"])
        self.encoder_optimizer.load_state_dict(states["encoder_opt"])
        self.decoder_optimizer.load_state_dict(states["decoder_opt"])
        self.frame_predictor_optimizer.load_state_dict(states["frame_predictor_opt"])
        self.posterior_optimizer.load_state_dict(states["posterior_opt"])
        self.prior_optimizer.load_state_dict(states["prior_opt"])
        return states["epoch"] + 1
