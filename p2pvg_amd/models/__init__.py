from .p2p import P2PModel, init_weights  # noqa: F401
from .lstm import lstm, gaussian_lstm  # noqa: F401
from . import backbones  # noqa: F401
