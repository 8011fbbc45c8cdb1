from .ddp import DDPGradSync  # noqa: F401
