from .distribution import Distribution  # noqa: F401
