from . import schema  # noqa: F401
