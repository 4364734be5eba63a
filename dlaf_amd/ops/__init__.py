from . import tile_ops  # noqa: F401
