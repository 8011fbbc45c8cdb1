"""Checkpoint round-trip: the 12-key dict contract."""
import argparse
import copy

import numpy as np
import torch

from p2pvg_amd.core import Config
from p2pvg_amd.models import P2PModel
from p2pvg_amd.utils import config_from_states, load_checkpoint, save_checkpoint

KEYS = {
    "encoder", "decoder", "frame_predictor", "posterior", "prior",
    "encoder_opt", "decoder_opt", "frame_predictor_opt", "posterior_opt",
    "prior_opt", "epoch", "opt",
}


def test_checkpoint_keys_and_roundtrip(tiny_cfg, tmp_path):
    torch.manual_seed(0)
    np.random.seed(0)
    model = P2PModel(tiny_cfg)
    # one step so optimizer state is non-trivial
    x = torch.rand(tiny_cfg.max_seq_len, tiny_cfg.batch_size, 1, 64, 64)
    model(x, 0, len(x) - 1)

    f = tmp_path / "model_0.pth"
    latest = tmp_path / "model.pth"
    save_checkpoint(model, 0, str(f), latest_name=str(latest))
    assert f.exists() and latest.exists()

    states = load_checkpoint(str(f))
    assert set(states.keys()) == KEYS
    assert isinstance(states["opt"], dict)

    cfg2 = config_from_states(states)
    model2 = P2PModel(cfg2)
    start_epoch = model2.load(states=states)
    assert start_epoch == 1
    for (n1, p1), (n2, p2) in zip(model.named_parameters(), model2.named_parameters()):
        assert torch.equal(p1, p2), n1


def test_namespace_opt_loads(tiny_cfg, tmp_path):
    """A reference-style checkpoint pickles an argparse.Namespace as `opt` —
    our loader must accept it (reference models/p2p_model.py:303)."""
    model = P2PModel(tiny_cfg)
    states = model.state_for_checkpoint(4)
    states["opt"] = argparse.Namespace(**tiny_cfg.to_dict())
    f = tmp_path / "ref_style.pth"
    torch.save(states, f)
    loaded = load_checkpoint(str(f))
    cfg = config_from_states(loaded)
    assert cfg.g_dim == tiny_cfg.g_dim
    model2 = P2PModel(cfg)
    assert model2.load(states=loaded) == 5


def test_config_dict_roundtrip(tiny_cfg):
    d = tiny_cfg.to_dict()
    assert "optimizer" not in d
    cfg2 = Config.from_dict(d)
    assert cfg2.to_dict() == d
