"""Backbone registry.

The reference selects its backbone by importing a module and stashing the module
object into the config (reference train.py:146-161, a pattern that forces
checkpoint code to null the field before pickling). We use a plain registry
keyed on (backbone, image_width, dataset) returning (encoder_cls, decoder_cls).
"""
from __future__ import annotations

from typing import Tuple, Type

from . import dcgan, h36m_mlp, vgg

_REGISTRY = {
    ("dcgan", 64): (dcgan.Encoder64, dcgan.Decoder64),
    ("dcgan", 128): (dcgan.Encoder128, dcgan.Decoder128),
    ("vgg", 64): (vgg.Encoder64, vgg.Decoder64),
    ("vgg", 128): (vgg.Encoder128, vgg.Decoder128),
    ("mlp", None): (h36m_mlp.Encoder, h36m_mlp.Decoder),
}


def get_backbone(backbone: str, image_width: int, dataset: str) -> Tuple[Type, Type]:
    if dataset == "h36m" or backbone == "mlp":
        return _REGISTRY[("mlp", None)]
    key = (backbone, image_width)
    if key not in _REGISTRY:
        raise ValueError(
            f"Unknown backbone {backbone!r} at image_width={image_width} "
            f"(known: {sorted(k for k in _REGISTRY if k[1])})"
        )
    return _REGISTRY[key]


def build_backbone(cfg):
    """Instantiate (encoder, decoder) from a Config."""
    enc_cls, dec_cls = get_backbone(cfg.backbone, cfg.image_width, cfg.dataset)
    if cfg.dataset == "h36m" or cfg.backbone == "mlp":
        encoder = enc_cls(out_dim=cfg.g_dim, h_dim=cfg.g_dim)
        decoder = dec_cls(in_dim=cfg.g_dim, h_dim=cfg.g_dim)
    else:
        encoder = enc_cls(cfg.g_dim, cfg.channels)
        decoder = dec_cls(cfg.g_dim, cfg.channels)
    return encoder, decoder
