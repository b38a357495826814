from . import common, config  # noqa: F401
