"""Functional alignment: SRM family.

Citation context (same methods as the reference implements):
[Chen2015] "A Reduced-Dimension fMRI Shared Response Model", NIPS 2015.
[Anderson2016] "Enabling Factor Analysis on Thousand-Subject Neuroimaging
Datasets", IEEE Big Data 2016.
"""

from .fastsrm import FastSRM  # noqa: F401
from .rsrm import RSRM  # noqa: F401
from .srm import SRM, DetSRM, load  # noqa: F401
from .sssrm import SSSRM  # noqa: F401

__all__ = ["FastSRM", "RSRM", "SRM", "SSSRM", "DetSRM", "load"]
