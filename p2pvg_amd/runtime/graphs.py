"""hipGraph capture of the whole training step (SURVEY §5.7 MI355X plan).

The per-timestep recurrence launches thousands of small kernels per training
step; capturing the entire step — forward over all processed timesteps, both
backward phases, and all five (capturable) Adam steps — into one hipGraph and
replaying it removes every launch gap and all host-side dispatch.

Dynamic control flow is handled exactly as SURVEY §5.7 prescribes:
- The skip gate and dynamic sequence length are HOST-side (plan_step); the
  captured compute depends only on the plan's graph_key (seq_len, n_proc,
  unpack pattern). One graph is captured per distinct key, on demand.
- Which frames feed each step, and the two time-signal scalars, are GRAPH
  INPUTS: the gathered (n,B,C,H,W) frame buffers and (n,1,1) scalar buffers
  are static tensors refilled before each replay.
- Reparameterization randomness uses the device philox generator, which
  torch.cuda.CUDAGraph captures/advances correctly.

Memory-pool discipline: every graph gets its OWN memory pool. Sharing one
pool across independently-replayed graphs corrupts state when replays happen
out of capture order (observed as negative MSE in long dynamic-length runs).
Since each pool pins a full step's activations, the number of live graphs is
capped (`max_graphs`); rarer (seq_len, skip-pattern) keys run eagerly through
the same model code path.

Capture warmup runs the step function a few times on a side stream (the
standard recipe: materializes grads, Adam state, autocast weight casts) —
those warmup iterations ARE real optimizer steps; for benchmarking they land
in the warmup phase, for training they are ordinary extra steps on the first
batch of that shape.
"""
from __future__ import annotations

import contextlib
from typing import Dict, Optional, Tuple

import torch

from ..models.p2p import StepPlan, gather_frames


class _GraphEntry:
    __slots__ = ("graph", "prev_buf", "cur_buf", "tun", "dts", "loss_out")


class GraphedTrainStep:
    def __init__(self, model, amp_dtype: Optional[torch.dtype] = None,
                 warmup_iters: int = 0, max_graphs: int = 16):
        self.model = model
        self.amp_dtype = amp_dtype
        self.warmup_iters = warmup_iters
        self.max_graphs = max_graphs
        self.graphs: Dict[Tuple, _GraphEntry] = {}
        self.eager_keys = set()
        self._state_ready = False

    def _materialize_persistent_state(self):
        """Allocate every step-persistent tensor (grad buffers, Adam state,
        conv weight shadows) BEFORE any warmup runs. Without this, the first
        warmup interleaves persistent allocations with a full step's
        transient activations, and the resulting fragmentation (persistent
        tensors pinning partially-used segments) was the vgg_128 batch-128
        capture OOM: 116 GB reserved-but-unallocated that the graph's
        private pool cannot reuse."""
        if self._state_ready:
            return
        model = self.model
        model.zero_grads()
        from ..optim import HIPFusedAdam

        for _, opt in self._optimizers():
            if isinstance(opt, HIPFusedAdam):
                for group in opt.param_groups:
                    for p in group["params"]:
                        state = opt.state[p]
                        if len(state) == 0:
                            state["step"] = torch.zeros(
                                (), dtype=torch.float32, device=p.device)
                            state["exp_avg"] = torch.zeros_like(p)
                            state["exp_avg_sq"] = torch.zeros_like(p)
        try:
            from ..ops.conv import refresh_conv_shadows

            refresh_conv_shadows(model)
        except Exception:  # noqa: BLE001 — shadows only exist on the HIP path
            pass
        self._state_ready = True

    def _amp_ctx(self, cache_enabled: bool):
        if self.amp_dtype is not None:
            return torch.autocast("cuda", dtype=self.amp_dtype,
                                  cache_enabled=cache_enabled)
        return contextlib.nullcontext()

    def _inner(self, entry: _GraphEntry, plan: StepPlan):
        model = self.model
        model.zero_grad(set_to_none=False)
        with self._amp_ctx(cache_enabled=False):
            losses = model._compute_losses(
                entry.prev_buf, entry.cur_buf, entry.tun, entry.dts, plan
            )
        model._backward_and_step(*losses)
        mse, kld, cpc, align = losses
        entry.loss_out.copy_(
            torch.stack([mse.detach(), kld.detach(), cpc.detach(), align.detach()])
        )

    def _eager_step(self, plan, prev, cur, tun, dts):
        model = self.model
        model.zero_grad(set_to_none=False)
        with self._amp_ctx(cache_enabled=True):
            losses = model._compute_losses(prev, cur, tun, dts, plan)
        model._backward_and_step(*losses)
        return tuple(v.detach() for v in losses)

    def _optimizers(self):
        return [v for v in vars(self.model).items()
                if isinstance(v[1], torch.optim.Optimizer)]

    def _snapshot_train_state(self):
        """Clone params/buffers + optimizer state so capture warmup applies
        zero net optimizer steps (warmup repeats the same batch several
        times; without the restore every new graph key would push up to
        warmup_iters duplicate-data updates into training)."""
        model_snap = {k: v.detach().clone()
                      for k, v in self.model.state_dict().items()}
        opt_snap = {}
        for name, opt in self._optimizers():
            opt_snap[name] = {
                id(p): {k: (v.detach().clone() if torch.is_tensor(v) else v)
                        for k, v in st.items()}
                for p, st in opt.state.items()
            }
        return model_snap, opt_snap

    def _restore_train_state(self, snap):
        """Copy the snapshot back IN PLACE (same storages: Adam state created
        during warmup stays allocated outside the graph pool; entries that
        did not exist at snapshot time are reset to the fresh-state zeros)."""
        model_snap, opt_snap = snap
        self.model.load_state_dict(model_snap)
        for name, opt in self._optimizers():
            saved = opt_snap.get(name, {})
            for p, st in opt.state.items():
                sv = saved.get(id(p))
                for k, v in st.items():
                    if not torch.is_tensor(v):
                        if sv is not None and k in sv:
                            st[k] = sv[k]
                        continue
                    if sv is not None and k in sv:
                        v.copy_(sv[k])
                    else:
                        v.zero_()

    def _dry_run(self, entry: _GraphEntry, plan: StepPlan):
        """One eager step at a TINY batch (2 samples) on a side stream.

        Materializes every lazy one-time initialization (hipBLASLt handles
        and workspaces, philox state, extension statics) WITHOUT running a
        full-size eager step first: the full-size warmup's ~20k transient
        allocations fragmented the caching allocator so badly (observed
        135 GB reserved-but-unallocated) that the subsequent capture pool
        OOM'd and the whole bench fell back to eager."""
        tiny = _GraphEntry()
        tiny.prev_buf = entry.prev_buf[:, :2].contiguous()
        tiny.cur_buf = entry.cur_buf[:, :2].contiguous()
        tiny.tun = entry.tun.clone()
        tiny.dts = entry.dts.clone()
        tiny.loss_out = torch.zeros(4, device=entry.loss_out.device)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._inner(tiny, plan)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

    def _capture(self, plan: StepPlan, prev: torch.Tensor, cur: torch.Tensor,
                 tun: torch.Tensor, dts: torch.Tensor) -> _GraphEntry:
        entry = _GraphEntry()
        entry.prev_buf = prev.clone()
        entry.cur_buf = cur.clone()
        entry.tun = tun.clone()
        entry.dts = dts.clone()
        entry.loss_out = torch.zeros(4, device=prev.device)

        self._materialize_persistent_state()
        snap = self._snapshot_train_state()
        if not getattr(self, "_lazy_init_done", False):
            self._dry_run(entry, plan)
            self._lazy_init_done = True
        # optional full-size warmup on a side stream (the standard CUDAGraph
        # recipe; persistent state is already materialized, so default 0)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self.warmup_iters):
                self._inner(entry, plan)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._restore_train_state(snap)
        del snap
        # release warmup's cached segments: expandable_segments is a no-op on
        # this ROCm build, and the graph's private pool cannot reuse the
        # allocator's fragmented free blocks (observed: 116 GB reserved but
        # unallocated OOM'ing a vgg_128 batch-128 capture)
        torch.cuda.empty_cache()

        g = torch.cuda.CUDAGraph()
        # private pool per graph: see the module docstring
        with torch.cuda.graph(g):
            self._inner(entry, plan)
        entry.graph = g
        return entry

    def step(self, x: torch.Tensor):
        """One training step on x ((T,B,...) device tensor or h36m tuple)."""
        model = self.model
        if isinstance(x, tuple):
            x = x[1]
        seq_len = len(x)
        plan = model.plan_step(seq_len)
        device = x.device

        idx_cur = torch.tensor(plan.proc, device=device)
        prev = gather_frames(x, idx_cur - 1)
        cur = gather_frames(x, idx_cur)
        tun = torch.as_tensor(plan.tun).to(device).view(-1, 1, 1)
        dts = torch.as_tensor(plan.dts).to(device).view(-1, 1, 1)

        key = plan.graph_key
        entry = self.graphs.get(key)
        if (entry is None and key not in self.eager_keys
                and len(self.graphs) < self.max_graphs):
            # capture records but does not execute; falls through to replay
            try:
                entry = self._capture(plan, prev, cur, tun, dts)
                self.graphs[key] = entry
            except (torch.OutOfMemoryError, RuntimeError) as e:
                import sys
                import traceback

                print(f"[graphs] capture failed for key {key}: {e!r} — "
                      "running this shape eagerly", file=sys.stderr)
                traceback.print_exc(file=sys.stderr)
                # capture needs headroom beyond the eager peak (side-stream
                # warmup segments + the graph's private pool cannot share
                # the allocator's fragmented free blocks, and
                # expandable_segments is a no-op on this ROCm build).
                # Large-batch configs train eagerly instead of not at all.
                # The fallback itself runs OUTSIDE this handler: the
                # exception's traceback pins every warmup frame (and so all
                # of that step's activations) until the handler exits.
                entry = None
        if entry is None:
            if key not in self.eager_keys:
                self.eager_keys.add(key)
                torch.cuda.empty_cache()
            losses = self._eager_step(plan, prev, cur, tun, dts)
            inv = 1.0 / seq_len
            return tuple(v * inv for v in losses)

        entry.prev_buf.copy_(prev)
        entry.cur_buf.copy_(cur)
        entry.tun.copy_(tun)
        entry.dts.copy_(dts)
        entry.graph.replay()
        inv = 1.0 / seq_len
        return tuple((entry.loss_out[i] * inv).clone() for i in range(4))
