from . import checkpoint, flat, metrics  # noqa: F401
