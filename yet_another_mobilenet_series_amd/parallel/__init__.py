from . import ddp, distributed  # noqa: F401
from .ddp import DataParallelEngine  # noqa: F401
