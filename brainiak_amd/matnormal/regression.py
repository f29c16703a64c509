"""Matrix-normal regression (API parity: ref
src/brainiak/matnormal/regression.py:15-146): maximum-likelihood
regression with structured temporal AND spatial residual covariance,
optimized by scipy L-BFGS over a torch-autograd value-and-gradient."""

import numpy as np
import torch
from scipy.optimize import minimize

from .matnormal_likelihoods import matnorm_logp
from .utils import make_val_and_grad, pack_trainable_vars

__all__ = ["MatnormalRegression"]

_DT = torch.float64


class MatnormalRegression:
    """Y ~ MN(Xβ, time_cov, space_cov); see module docstring."""

    def __init__(self, time_cov, space_cov, optimizer="L-BFGS-B",
                 optCtrl=None, device=None):
        self.optMethod = optimizer
        self.optCtrl = optCtrl if optCtrl is not None else {}
        self.device = torch.device(device) if device is not None \
            else torch.device("cpu")
        self.time_cov = time_cov.to(self.device)
        self.space_cov = space_cov.to(self.device)
        self.n_t = time_cov.size
        self.n_v = space_cov.size

    def logp(self, X, Y):
        resid = Y - X @ self.beta
        return matnorm_logp(resid, self.time_cov, self.space_cov)

    def fit(self, X, y, naive_init=True):
        """Fit β and covariance params to design X [T, C], data y [T, V]."""
        X = torch.as_tensor(np.asarray(X), dtype=_DT,
                            device=self.device)
        y = torch.as_tensor(np.asarray(y), dtype=_DT,
                            device=self.device)
        self.n_c = X.shape[1]

        if naive_init:
            with torch.no_grad():
                sigma_inv_x = self.time_cov.solve(X)
                sigma_inv_y = self.time_cov.solve(y)
                beta_init = torch.linalg.solve(X.T @ sigma_inv_x,
                                               X.T @ sigma_inv_y)
        else:
            beta_init = torch.randn((self.n_c, self.n_v), dtype=_DT,
                                    device=self.device)
        self.beta = beta_init.clone().detach().requires_grad_(True)

        self.train_variables = [self.beta]
        self.train_variables.extend(self.time_cov.get_optimize_vars())
        self.train_variables.extend(self.space_cov.get_optimize_vars())

        def lossfn(theta):
            return -(self.logp(X, y)
                     + self.time_cov.logp + self.space_cov.logp)

        val_and_grad = make_val_and_grad(lossfn, self.train_variables)
        x0 = pack_trainable_vars(self.train_variables).numpy()
        opt_results = minimize(fun=val_and_grad, x0=x0, jac=True,
                               method=self.optMethod, **self.optCtrl)
        from .utils import unpack_trainable_vars
        unpacked = unpack_trainable_vars(opt_results.x,
                                         self.train_variables)
        for var, val in zip(self.train_variables, unpacked):
            with torch.no_grad():
                var.copy_(val)
        self.beta_ = self.beta.detach().cpu().numpy()
        return self

    def predict(self, X):
        return np.asarray(X).dot(self.beta_)

    def calibrate(self, Y):
        """MLE decode of the design from data given the fitted β."""
        if Y.shape[1] <= self.n_c:
            raise RuntimeError(
                "More conditions than voxels! System is singular, "
                "cannot decode.")
        Y = torch.as_tensor(np.asarray(Y), dtype=_DT,
                            device=self.device)
        beta = torch.as_tensor(self.beta_, dtype=_DT,
                               device=self.device)
        with torch.no_grad():
            Sigma_s_btrp = self.space_cov.solve(beta.T)
            Y_Sigma_Btrp = (Y @ Sigma_s_btrp).cpu().numpy()
            B_Sigma_Btrp = (beta @ Sigma_s_btrp).cpu().numpy()
        return np.linalg.solve(B_Sigma_Btrp.T, Y_Sigma_Btrp.T).T
