"""The pruned two-phase backward must produce identical parameter updates to
the reference's literal double full-graph backward (SURVEY §2.2 semantics)."""
import copy

import numpy as np
import pytest
import torch

from p2pvg_amd.core import Config
from p2pvg_amd.models import P2PModel


def _make(cfg):
    torch.manual_seed(7)
    model = P2PModel(cfg)
    return model


def _step(model, x, mode, seed):
    model.cfg.backward_mode = mode
    torch.manual_seed(seed)
    np.random.seed(seed)
    model.zero_grad(set_to_none=False)
    return model(x, 0, len(x) - 1)


@pytest.mark.parametrize("align_mode", ["paper", "reference"])
def test_pruned_backward_matches_reference(tiny_cfg, align_mode):
    tiny_cfg.weight_align = 0.5
    tiny_cfg.align_mode = align_mode
    tiny_cfg.skip_prob = 0.3

    m1 = _make(tiny_cfg)
    m2 = copy.deepcopy(m1)

    g = torch.Generator().manual_seed(3)
    x = torch.rand(
        tiny_cfg.max_seq_len, tiny_cfg.batch_size, 1, 64, 64, generator=g
    )

    l1 = _step(m1, x, "reference", seed=11)
    l2 = _step(m2, x, "pruned", seed=11)

    for a, b in zip(l1, l2):
        assert torch.allclose(a, b, atol=1e-6), f"losses differ: {a} vs {b}"

    for (n1, p1), (n2, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert n1 == n2
        assert torch.allclose(p1, p2, rtol=1e-5, atol=1e-7), (
            f"param {n1} diverged: max abs diff "
            f"{(p1 - p2).abs().max().item()}"
        )
