from .logging import ScalarWriter, get_logger, store_cmd  # noqa: F401
from .checkpoint import config_from_states, load_checkpoint, save_checkpoint  # noqa: F401
from .metrics import Metric, end_frame_ssim, mse, psnr, ssim  # noqa: F401
