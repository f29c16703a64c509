"""Matrix-normal RSA (API parity: ref src/brainiak/matnormal/
mnrsa.py:21-175): marginalizes over the design→signal mapping to
de-bias the RSA covariance estimate (Cai et al., NIPS 2016), with
Kronecker-separable residual covariance."""

import numpy as np
import torch
from scipy.optimize import minimize

from ..utils.utils import cov2corr
from .covs import CovIdentity
from .matnormal_likelihoods import matnorm_logp_marginal_row
from .utils import (
    flatten_cholesky_unique,
    make_val_and_grad,
    pack_trainable_vars,
    unflatten_cholesky_unique,
    unpack_trainable_vars,
)

__all__ = ["MNRSA"]

_DT = torch.float64


class MNRSA:
    """Y ~ MN(0, Σ_t + XLLᵀXᵀ + X₀X₀ᵀ, Σ_s), U = LLᵀ; see docstring."""

    def __init__(self, time_cov, space_cov, n_nureg=5,
                 optimizer="L-BFGS-B", optCtrl=None, device=None):
        self.n_T = time_cov.size
        self.n_V = space_cov.size
        self.n_nureg = n_nureg
        self.optMethod = optimizer
        self.optCtrl = optCtrl if optCtrl is not None else {}
        self.device = torch.device(device) if device is not None \
            else torch.device("cpu")
        self.X_0 = torch.randn((self.n_T, n_nureg), dtype=_DT,
                               device=self.device).requires_grad_(True)
        self.train_variables = [self.X_0]
        self.time_cov = time_cov.to(self.device)
        self.space_cov = space_cov.to(self.device)
        self.train_variables.extend(self.time_cov.get_optimize_vars())
        self.train_variables.extend(self.space_cov.get_optimize_vars())

    @property
    def L(self):
        return unflatten_cholesky_unique(self.L_flat)

    def fit(self, X, y, naive_init=True):
        """X = brain data [T, V], y = design [T, C] (sklearn-style
        argument order, flipped internally like the reference)."""
        X_design, Y_brain = y, X
        X_design = torch.as_tensor(np.asarray(X_design), dtype=_DT,
                                   device=self.device)
        Y_brain = torch.as_tensor(np.asarray(Y_brain), dtype=_DT,
                                  device=self.device)
        self.n_c = X_design.shape[1]

        if naive_init:
            from sklearn.linear_model import LinearRegression
            m = LinearRegression(fit_intercept=False)
            m.fit(X=X_design.cpu().numpy(), y=Y_brain.cpu().numpy())
            self.naive_U_ = np.cov(m.coef_.T)
            # regularize in case the naive estimate is singular
            naive = self.naive_U_ + 1e-9 * np.eye(self.n_c)
            self.L_flat = flatten_cholesky_unique(
                np.linalg.cholesky(naive)).clone().to(
                    self.device).requires_grad_(True)
        else:
            chol_flat_size = (self.n_c * (self.n_c + 1)) // 2
            self.L_flat = torch.randn(
                chol_flat_size, dtype=_DT,
                device=self.device).requires_grad_(True)
        self.train_variables = self.train_variables + [self.L_flat]

        def lossfn(theta):
            return -self.logp(X_design, Y_brain)

        val_and_grad = make_val_and_grad(lossfn, self.train_variables)
        x0 = pack_trainable_vars(self.train_variables).numpy()
        opt_results = minimize(fun=val_and_grad, x0=x0, jac=True,
                               method=self.optMethod, **self.optCtrl)
        unpacked = unpack_trainable_vars(opt_results.x,
                                         self.train_variables)
        for var, val in zip(self.train_variables, unpacked):
            with torch.no_grad():
                var.copy_(val)
        L = self.L.detach().cpu().numpy()
        self.U_ = L.dot(L.T)
        self.C_ = cov2corr(self.U_)
        return self

    def logp(self, X, Y):
        """MNRSA log-likelihood (marginal over the mapping)."""
        rsa_cov = CovIdentity(size=self.n_c + self.n_nureg).to(
            self.device)
        x_stack = torch.cat([X @ self.L, self.X_0], dim=1)
        return (self.time_cov.logp + self.space_cov.logp + rsa_cov.logp
                + matnorm_logp_marginal_row(
                    Y, row_cov=self.time_cov, col_cov=self.space_cov,
                    marg=x_stack, marg_cov=rsa_cov))
