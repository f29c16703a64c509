from .searchlight import Ball, Cube, Diamond, Searchlight, Shape  # noqa: F401

__all__ = ["Ball", "Cube", "Diamond", "Searchlight", "Shape"]
