from .builder import Builder  # noqa: F401
