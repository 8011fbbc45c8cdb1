"""RCCL-over-xGMI data parallelism for the two-phase P2P update.

The reference is single-GPU only — this component is new,
designed for one process per MI355X GPU with torch.distributed's "nccl"
backend (= RCCL on ROCm) over xGMI point-to-point links.

Design:
- Two distinct flush points per iteration, matching the two-phase update:
  grads of {encoder, decoder, frame_predictor, posterior} are all-reduced
  after backward phase 1 / before their Adam steps, and the prior's grads
  after backward phase 2 / before the prior step. P2PModel calls
  `sync_nonprior()` / `sync_prior()` when `model.grad_sync` is set.
- Buckets: xGMI ring all-reduce is per-link bound (7 x ~153 GB/s), so the many
  tiny LSTM/Linear tensors are coalesced into flat buckets of ~bucket_mb MB;
  reduction of bucket k overlaps the DtoD gather of bucket k+1 on the main
  stream (async_op handles).
- Gradient averaging divides by world size after the sum-reduce.
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist
from torch._utils import _flatten_dense_tensors, _unflatten_dense_tensors


class DDPGradSync:
    def __init__(self, model, bucket_mb: int = 50, process_group=None):
        self.model = model
        self.group = process_group
        self.world = dist.get_world_size(process_group)
        self.bucket_bytes = bucket_mb * 1024 * 1024
        nonprior = [
            p
            for n in model.NONPRIOR
            for p in getattr(model, n).parameters()
            if p.requires_grad
        ]
        prior = [p for p in model.prior.parameters() if p.requires_grad]
        self.nonprior_buckets = self._partition(nonprior)
        self.prior_buckets = self._partition(prior)
        model.grad_sync = self

    def _partition(self, params: List[torch.Tensor]) -> List[List[torch.Tensor]]:
        buckets, cur, cur_bytes = [], [], 0
        # reverse order: parameters produced late in backward reduce first
        for p in reversed(params):
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= self.bucket_bytes:
                buckets.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            buckets.append(cur)
        return buckets

    def broadcast_parameters(self) -> None:
        """Rank-0 weights (and Adam state, on resume) to all ranks."""
        for p in self.model.parameters():
            dist.broadcast(p.data, src=0, group=self.group)
        for b in self.model.buffers():
            dist.broadcast(b.data, src=0, group=self.group)

    def _sync_buckets(self, buckets: List[List[torch.Tensor]]) -> None:
        if self.world <= 1:
            return
        inflight = []
        for bucket in buckets:
            grads = [p.grad for p in bucket if p.grad is not None]
            if not grads:
                continue
            flat = _flatten_dense_tensors(grads)
            work = dist.all_reduce(flat, group=self.group, async_op=True)
            inflight.append((work, flat, grads))
        inv = 1.0 / self.world
        for work, flat, grads in inflight:
            work.wait()
            flat.mul_(inv)
            for g, synced in zip(grads, _unflatten_dense_tensors(flat, grads)):
                g.copy_(synced)

    def sync_nonprior(self) -> None:
        self._sync_buckets(self.nonprior_buckets)

    def sync_prior(self) -> None:
        self._sync_buckets(self.prior_buckets)

    def all_reduce_scalars(self, t: torch.Tensor) -> torch.Tensor:
        if self.world <= 1:
            return t
        dist.all_reduce(t, group=self.group)
        return t / self.world
