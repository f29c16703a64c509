from .event import EventSegment  # noqa: F401

__all__ = ["EventSegment"]
