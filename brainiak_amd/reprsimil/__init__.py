"""Bayesian representational similarity analysis.

Citation: [Cai2016] "A Bayesian method for reducing bias in neural
representational similarity analysis", NIPS 2016; [Cai2019] extended
version in PLoS Computational Biology 15(5).
"""

from .brsa import BRSA, GBRSA, Ncomp_SVHT_MG_DLD_approx  # noqa: F401

__all__ = ["BRSA", "GBRSA", "Ncomp_SVHT_MG_DLD_approx"]
