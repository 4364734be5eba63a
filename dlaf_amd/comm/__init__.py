from .grid import CommGrid  # noqa: F401
