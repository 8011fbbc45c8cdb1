"""Multi-process CPU test of the DDP gradient sync (gloo, world_size=2).

Checks that after sync_nonprior/sync_prior, every rank holds the average of
per-rank gradients — i.e. a 2-rank step on batches [A|B] equals a single-rank
step on the concatenated batch (same loss normalization caveats aside).
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from p2pvg_amd.core import Config


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from p2pvg_amd.models import P2PModel
        from p2pvg_amd.parallel import DDPGradSync

        cfg = Config(dataset="mnist", backbone="dcgan", batch_size=2, max_seq_len=6,
                     g_dim=16, z_dim=4, rnn_size=32, device="cpu", skip_prob=0.0,
                     lr=0.0)  # lr=0: isolate gradient averaging from updates
        torch.manual_seed(42)  # same init on both ranks
        model = P2PModel(cfg)
        sync = DDPGradSync(model, bucket_mb=1)
        sync.broadcast_parameters()

        torch.manual_seed(100 + rank)
        np.random.seed(100 + rank)
        x = torch.rand(6, 2, 1, 64, 64)
        model.zero_grad(set_to_none=False)
        model(x, 0, 5)

        # after the model step, grads have been synced (sync runs inside
        # _backward_and_step); collect a fingerprint
        fp = []
        for p in model.parameters():
            if p.grad is not None:
                fp.append(p.grad.sum().item())
        results[rank] = fp
    finally:
        dist.destroy_process_group()


def _single_rank_grads(rank):
    """Replicate one rank's step WITHOUT distributed: same init (seed 42),
    same data/plan seeds (100+rank); returns the per-param grad sums."""
    from p2pvg_amd.models import P2PModel

    cfg = Config(dataset="mnist", backbone="dcgan", batch_size=2, max_seq_len=6,
                 g_dim=16, z_dim=4, rnn_size=32, device="cpu", skip_prob=0.0,
                 lr=0.0)
    torch.manual_seed(42)
    model = P2PModel(cfg)
    torch.manual_seed(100 + rank)
    np.random.seed(100 + rank)
    x = torch.rand(6, 2, 1, 64, 64)
    model.zero_grad(set_to_none=False)
    model(x, 0, 5)
    return [p.grad.sum().item() for p in model.parameters() if p.grad is not None]


@pytest.mark.parametrize("world", [2, 4])
def test_ddp_grad_sync(world):
    port = 29511 + world
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_worker, args=(r, world, port, results))
            for r in range(world)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
        for p in procs:
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
        fps = [results[r] for r in range(world)]
    assert all(len(fp) == len(fps[0]) for fp in fps) and len(fps[0]) > 0
    for vals in zip(*fps):
        for v in vals[1:]:
            assert v == pytest.approx(vals[0], rel=1e-5, abs=1e-7), "grads not synced"

    # the synced value must BE the average of the per-rank gradients
    # (replicated serially without torch.distributed): a wrong all-reduce
    # (sum instead of mean, wrong bucket unflatten) fails here
    per_rank = [_single_rank_grads(r) for r in range(world)]
    for i, vals in enumerate(zip(*per_rank)):
        want = sum(vals) / world
        got = fps[0][i]
        assert got == pytest.approx(want, rel=1e-4, abs=1e-6), (
            f"param {i}: synced {got} != mean {want}"
        )
