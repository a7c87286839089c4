from .ddp import DDPEngine  # noqa: F401
