from . import utils  # noqa: F401

__all__ = ["utils", "fmrisim"]
