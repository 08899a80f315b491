from .net import Net  # noqa: F401
