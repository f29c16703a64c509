from . import hpo  # noqa: F401

__all__ = ["hpo"]
