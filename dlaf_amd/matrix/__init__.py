from .matrix import Matrix  # noqa: F401
