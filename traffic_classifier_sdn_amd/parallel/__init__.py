from . import dist  # noqa: F401
