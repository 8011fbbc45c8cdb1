"""Fused Conv -> BatchNorm -> activation path (SURVEY §2.6 K1+K4+K5).

The conv kernel's epilogue accumulates per-channel sum/sumsq, so the whole
block costs: conv (one kernel) + BN finalize (K threads) + one elementwise
normalize+activation pass. Backward folds the activation grad into the BN
reduction/apply kernels. `FusedSequential` pattern-matches the reference's
Sequential(conv, bn, act) blocks (state_dict keys unchanged) and routes them
here on the gfx950 path.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

CL = torch.channels_last


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


ACT_NONE, ACT_LEAKY, ACT_TANH, ACT_SIGMOID = 0, 1, 2, 3


def act_code(mod: nn.Module) -> Optional[int]:
    if isinstance(mod, nn.LeakyReLU) and abs(mod.negative_slope - 0.2) < 1e-8:
        return ACT_LEAKY
    if isinstance(mod, nn.Tanh):
        return ACT_TANH
    if isinstance(mod, nn.Sigmoid):
        return ACT_SIGMOID
    return None


class FusedBNActFn(torch.autograd.Function):
    """y = act(bn(x)) with batch statistics from a pre-accumulated (2,C)
    sum/sumsq buffer (conv epilogue) and fused activation backward."""

    @staticmethod
    def forward(ctx, x, stats, gamma, beta, running_mean, running_var,
                momentum: float, eps: float, act: int, ring: int = 0):
        # ring > 0: x/y are padded maps — the kernels walk the interior and
        # write y's ring as ZEROS (it becomes the next conv's padding)
        ext = _ext()
        gamma32 = gamma.float()
        beta32 = beta.float()
        y, mean, invstd, scale = ext.bn_act_fwd_train(
            x, stats, gamma32, beta32, running_mean, running_var,
            momentum, eps, act, ring,
        )
        ctx.save_for_backward(x, mean, invstd, gamma32, beta32, scale)
        ctx.act = act
        ctx.ring = ring
        ctx.gref, ctx.bref = gamma, beta
        return y

    @staticmethod
    def backward(ctx, dy):
        from .conv import _acc_target, weight_grads_enabled

        x, mean, invstd, gamma32, beta32, scale = ctx.saved_tensors
        ext = _ext()
        dy = dy.contiguous(memory_format=CL)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        r = ctx.ring
        if not weight_grads_enabled(ctx.gref):
            # phase-2 traversal: dx only, param grads are discarded anyway
            dx, _, _ = ext.bn_act_bwd(
                x, dy, mean, invstd, gamma32, beta32, scale, ctx.act,
                None, None, r,
            )
            return dx, None, None, None, None, None, None, None, None, None
        ga = _acc_target(ctx.gref)
        ba = _acc_target(ctx.bref)
        if ga is not None and ba is not None:
            # accumulate dgamma/dbeta straight into the managed .grad buffers
            dx, _, _ = ext.bn_act_bwd(
                x, dy, mean, invstd, gamma32, beta32, scale, ctx.act, ga, ba, r
            )
            return dx, None, None, None, None, None, None, None, None, None
        dx, dgamma, dbeta = ext.bn_act_bwd(
            x, dy, mean, invstd, gamma32, beta32, scale, ctx.act, None, None, r
        )
        return dx, None, dgamma, dbeta, None, None, None, None, None, None


def fused_conv_bn_act(x, conv, bn, act: int, pad_out: bool = False):
    """conv (with stats epilogue) -> fused BN+act. Training and eval modes.

    pad_out: emit y as a PADDED map whose zero ring realizes the NEXT conv's
    padding (its gathers become in-bounds -> glds staging pipeline). An
    incoming padded x (attribute `_pvg_pad` set by the previous producer)
    routes the conv itself through the in-bounds path."""
    from .conv import (CatConv2dFn, Conv2dNHWCFn, ConvT2dNHWCFn,
                       _conv_shadows, _to_cl_bf16)

    ext = _ext()
    training = bn.training
    dual = isinstance(x, tuple)
    in_ring = getattr(x[0] if dual else x, "_pvg_pad", 0)
    out_ring = 1 if pad_out else 0
    with torch.autocast("cuda", enabled=False):
        sh = _conv_shadows(conv)
        want_stats = training
        # BatchNorm is shift-invariant, so the conv bias has EXACTLY zero
        # effect on the block output; skip it (and its gradient reduction).
        # The reference trains this bias on float-rounding noise only — it is
        # initialized to 0 (init_weights) and stays ~0.
        if dual:
            # skip-concat elimination: the conv gathers from both sources
            out, stats = CatConv2dFn.apply(
                _to_cl_bf16(x[0]), _to_cl_bf16(x[1]), conv.weight,
                conv.stride[0], conv.padding[0], 0, want_stats,
                sh["f"], sh["b"], in_ring, out_ring
            )
        elif isinstance(conv, nn.ConvTranspose2d):
            xl = _to_cl_bf16(x)
            out, stats = ConvT2dNHWCFn.apply(
                xl, conv.weight, None, conv.stride[0], conv.padding[0],
                0, want_stats, sh["f"], sh["b"], in_ring, out_ring
            )
        else:
            xl = _to_cl_bf16(x)
            out, stats = Conv2dNHWCFn.apply(
                xl, conv.weight, None, conv.stride[0], conv.padding[0],
                0, want_stats, sh["f"], sh["b"], in_ring, out_ring
            )
        if not want_stats:
            stats = None
        if training:
            if bn.track_running_stats and bn.num_batches_tracked is not None:
                # keep nn.BatchNorm2d state semantics (cumulative-average
                # momentum=None and checkpoint parity); device-side add_ stays
                # hipGraph-capturable
                bn.num_batches_tracked.add_(1)
            y = FusedBNActFn.apply(
                out, stats, bn.weight, bn.bias,
                bn.running_mean if bn.track_running_stats else None,
                bn.running_var if bn.track_running_stats else None,
                bn.momentum if bn.momentum is not None else 0.1,
                bn.eps, act, out_ring,
            )
        else:
            y = ext.bn_act_fwd_eval(
                out, bn.weight.float(), bn.bias.float(), bn.running_mean,
                bn.running_var, bn.eps, act, out_ring,
            )
        if out_ring:
            y._pvg_pad = out_ring
        return y


def fused_conv_act(x, conv, act: int):
    """conv with the activation fused straight into the epilogue (no BN).
    Output is always dense; a padded input (attr) is consumed in-bounds."""
    from .conv import Conv2dNHWCFn, ConvT2dNHWCFn, _conv_shadows, _to_cl_bf16

    in_ring = getattr(x, "_pvg_pad", 0)
    with torch.autocast("cuda", enabled=False):
        xl = _to_cl_bf16(x)
        sh = _conv_shadows(conv)
        if isinstance(conv, nn.ConvTranspose2d):
            out, _ = ConvT2dNHWCFn.apply(
                xl, conv.weight, conv.bias, conv.stride[0], conv.padding[0],
                act, False, sh["f"], sh["b"], in_ring, 0
            )
        else:
            out, _ = Conv2dNHWCFn.apply(
                xl, conv.weight, conv.bias, conv.stride[0], conv.padding[0],
                act, False, sh["f"], sh["b"], in_ring, 0
            )
        return out


def _conv_supported(conv, x) -> bool:
    if isinstance(conv, nn.ConvTranspose2d):
        extra = conv.output_padding == (0, 0)
    else:
        extra = True
    return (
        extra
        and conv.kernel_size[0] == conv.kernel_size[1]
        and conv.stride[0] == conv.stride[1]
        and conv.padding[0] == conv.padding[1]
        and conv.kernel_size[0] in (1, 2, 3, 4)
        and conv.stride[0] in (1, 2)
        and conv.dilation == (1, 1)
        and conv.groups == 1
    )


def _consumes_ring(m) -> bool:
    """Does module m's conv accept a padded (zero-ring) input in place of its
    own padding? True for the supported pad-1 geometries (k3s1p1, k4s2p1,
    and convT k3s1p1 whose flipped-conv pad is 1)."""
    if isinstance(m, nn.ConvTranspose2d):
        return (m.kernel_size[0] == 3 and m.stride[0] == 1
                and m.padding[0] == 1)
    if isinstance(m, nn.Conv2d):
        return m.padding[0] == 1 and m.kernel_size[0] in (3, 4)
    return False


class FusedSequential(nn.Sequential):
    """nn.Sequential that fuses (conv [, batchnorm] [, activation]) chains on
    the gfx950 path. Same children indices -> same state_dict keys. Greedy
    scan, so trailing conv+act pairs inside longer chains fuse too.

    pad_out (attr `_pvg_pad_out`, set by the backbone): emit the LAST fused
    block's output as a padded zero-ring map for the consumer conv outside
    this Sequential. Inside the chain, a block pads out whenever the next
    module's conv consumes a ring."""

    def __init__(self, *args, pad_out: bool = False):
        super().__init__(*args)
        self._pvg_pad_out = pad_out

    def forward(self, x):
        from .conv import _use_hip_path, cat_conv_eligible

        mods = list(self)
        i = 0
        while i < len(mods):
            m = mods[i]
            if isinstance(x, tuple):
                # dual-source pair from _skip_join: consumed by the fused
                # conv below, passed through to a block whose own inner
                # FusedSequential will consume it (vgg_layer and friends
                # forward to .main), or materialized as a plain concat
                fused_dual = (
                    i + 2 < len(mods)
                    and cat_conv_eligible(x[0], x[1], m)
                    and _use_hip_path(x[0])
                    and isinstance(mods[i + 1], nn.BatchNorm2d)
                    and act_code(mods[i + 2]) is not None
                    and mods[i + 1].num_features % 8 == 0
                    and 256 % (mods[i + 1].num_features // 8) == 0
                )
                inner = getattr(m, "main", None)
                if not fused_dual and not isinstance(inner, FusedSequential):
                    pa = getattr(x[0], "_pvg_pad", 0)
                    x = torch.cat(x, 1)
                    if pa:
                        x._pvg_pad = pa
            if (
                isinstance(x, tuple)
                or (isinstance(m, (nn.Conv2d, nn.ConvTranspose2d))
                    and _use_hip_path(x)
                    and _conv_supported(m, x))
            ):
                if (
                    i + 2 < len(mods)
                    and isinstance(mods[i + 1], nn.BatchNorm2d)
                    and act_code(mods[i + 2]) is not None
                    and mods[i + 1].num_features % 8 == 0
                    and 256 % (mods[i + 1].num_features // 8) == 0
                ):
                    if i + 3 < len(mods):
                        pad_out = _consumes_ring(mods[i + 3])
                    else:
                        pad_out = getattr(self, "_pvg_pad_out", False)
                    x = fused_conv_bn_act(x, m, mods[i + 1],
                                          act_code(mods[i + 2]), pad_out)
                    i += 3
                    continue
                if i + 1 < len(mods) and act_code(mods[i + 1]) is not None:
                    x = fused_conv_act(x, m, act_code(mods[i + 1]))
                    i += 2
                    continue
            x = m(x)
            i += 1
        return x
