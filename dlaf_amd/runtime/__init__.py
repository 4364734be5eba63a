from .streams import Runtime, get_runtime  # noqa: F401
