"""Topographic factor analysis (TFA) and its hierarchical multi-subject
extension (HTFA).

Citations (methods as in the reference): [Manning2014] "Topographic factor
analysis: a Bayesian model for inferring brain networks from neural data",
PLoS ONE 9(5); [AndersonMJ2016] "Scaling up multi-subject neuroimaging
factor analysis", arXiv 1608.04647.
"""

from .htfa import HTFA  # noqa: F401
from .tfa import TFA  # noqa: F401

__all__ = ["HTFA", "TFA"]
