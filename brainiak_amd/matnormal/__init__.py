"""Matrix-normal models (torch re-expression of the reference's
TF-based toolkit).  Citation: [Shvartsman2018] "Matrix-normal models
for fMRI analysis", AISTATS 2018."""

from .covs import (  # noqa: F401
    CovAR1,
    CovBase,
    CovDiagonal,
    CovDiagonalGammaPrior,
    CovIdentity,
    CovIsotropic,
    CovKroneckerFactored,
    CovUnconstrainedCholesky,
    CovUnconstrainedCholeskyWishartReg,
    CovUnconstrainedInvCholesky,
)
from .matnormal_likelihoods import (  # noqa: F401
    matnorm_logp,
    matnorm_logp_conditional_col,
    matnorm_logp_conditional_row,
    matnorm_logp_marginal_col,
    matnorm_logp_marginal_row,
)
from .mnrsa import MNRSA  # noqa: F401
from .regression import MatnormalRegression  # noqa: F401

__all__ = [
    "CovAR1",
    "CovBase",
    "CovDiagonal",
    "CovDiagonalGammaPrior",
    "CovIdentity",
    "CovIsotropic",
    "CovKroneckerFactored",
    "CovUnconstrainedCholesky",
    "CovUnconstrainedCholeskyWishartReg",
    "CovUnconstrainedInvCholesky",
    "MNRSA",
    "MatnormalRegression",
    "matnorm_logp",
    "matnorm_logp_conditional_col",
    "matnorm_logp_conditional_row",
    "matnorm_logp_marginal_col",
    "matnorm_logp_marginal_row",
]
