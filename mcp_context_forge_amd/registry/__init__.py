from .registry import Registry  # noqa: F401
