"""Typed, dict-serializable configuration for p2pvg_amd.

Replaces the reference's flat argparse Namespace (reference train.py:33-71) with a
dataclass that keeps the same flag names and defaults, serializes to/from a plain
dict (so checkpoints never pickle a Namespace or a module object — contrast
reference models/p2p_model.py:291-292 which must null `opt.backbone_net` before
torch.save), and adds the MI355X-native knobs (device, dtype, kernel backend,
hipGraph capture, DDP).
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class Config:
    # -- reference flags (names and defaults preserved; reference train.py:33-71) --
    seed: int = 1
    log_dir: str = "logs/p2pvg"
    data_root: str = "data_root"
    ckpt: str = ""
    dataset: str = "mnist"            # mnist | weizmann | h36m | bair
    num_digits: int = 1
    nepochs: int = 200
    epoch_size: int = 300
    lr: float = 0.001
    batch_size: int = 22
    beta1: float = 0.9
    image_width: int = 64
    channels: int = 1
    n_past: int = 1
    nsample: int = 20
    rnn_size: int = 256
    prior_rnn_layers: int = 1
    posterior_rnn_layers: int = 1
    predictor_rnn_layers: int = 2
    z_dim: int = 10
    g_dim: int = 128
    beta: float = 0.0001
    backbone: str = "dcgan"           # dcgan | vgg | mlp
    last_frame_skip: bool = False
    max_seq_len: int = 30
    delta_len: int = 5
    weight_cpc: float = 1000.0
    weight_align: float = 0.0
    skip_prob: float = 0.1
    qual_iter: int = 1
    quan_iter: int = 1
    test: bool = False

    # -- MI355X-native additions (no reference counterpart) --
    device: str = "auto"              # auto | cpu | cuda | cuda:N
    dtype: str = "fp32"               # fp32 | bf16  (bf16 = autocast compute, fp32 master/BN)
    kernels: str = "auto"             # auto | hip | torch  (hot-op backend dispatch)
    use_graphs: bool = False          # hipGraph-capture the per-timestep step body
    align_mode: str = "paper"         # paper | reference (reference = as-written h[0] broadcast,
                                      #   see reference models/p2p_model.py:225 and SURVEY §2.2)
    ddp: bool = False                 # RCCL data-parallel (one process per GPU)
    bucket_mb: int = 50               # gradient all-reduce bucket size (xGMI-link sized)
    log_histograms: bool = False      # per-param tensorboard histograms (reference does this
                                      #   unconditionally every 50 iters, a full DtoH sync)
    deterministic: bool = False
    num_workers: int = 1

    # -- runtime-derived (never serialized as live objects) --
    optimizer: Any = field(default=None, repr=False, compare=False)

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d.pop("optimizer", None)
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Config":
        names = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in names and k != "optimizer"})

    # argparse-Namespace compatibility: old code (and pickled reference checkpoints
    # re-read through our loader) accesses cfg.<flag> directly, which dataclass
    # attributes already provide.

    def resolved_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch
        return "cuda" if torch.cuda.is_available() else "cpu"


def add_config_args(parser) -> None:
    """Register every Config field as a CLI flag (reference-compatible names)."""
    import argparse

    for f in dataclasses.fields(Config):
        if f.name == "optimizer":
            continue
        if f.type == "bool" or isinstance(f.default, bool):
            parser.add_argument(f"--{f.name}", action="store_true", default=f.default)
        else:
            ftype = type(f.default) if f.default is not None else str
            parser.add_argument(f"--{f.name}", type=ftype, default=f.default)
    # reference also has --gpu (maps to device)
    parser.add_argument("--gpu", type=int, default=None,
                        help="gpu index to use (reference-compatible; sets device=cuda:N)")


def config_from_args(args) -> Config:
    d = vars(args).copy()
    gpu = d.pop("gpu", None)
    names = {f.name for f in dataclasses.fields(Config)}
    cfg = Config(**{k: v for k, v in d.items() if k in names and k != "optimizer"})
    if gpu is not None:
        cfg.device = f"cuda:{gpu}"
    return cfg
