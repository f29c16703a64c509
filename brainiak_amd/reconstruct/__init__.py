from .iem import InvertedEncoding1D, InvertedEncoding2D  # noqa: F401

__all__ = ["InvertedEncoding1D", "InvertedEncoding2D"]
