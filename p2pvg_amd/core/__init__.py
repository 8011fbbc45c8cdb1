from .config import Config, add_config_args, config_from_args

__all__ = ["Config", "add_config_args", "config_from_args"]
