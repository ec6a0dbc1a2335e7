from .ddp import DataParallelGroup  # noqa: F401
