"""Residual covariance classes for matrix-normal models.

Torch re-expression of the reference's TF covariance toolkit
(ref src/brainiak/matnormal/covs.py:35-660): each class exposes
``logdet``, ``solve(X)`` (Σ⁻¹X), ``get_optimize_vars()`` (torch
parameter tensors optimized by autograd + scipy L-BFGS), and an
optional regularization log-prob ``logp``.
"""

import abc

import numpy as np
import scipy.linalg
import scipy.sparse
import scipy.special
import torch

from ..utils.kronecker_solvers import (
    solve_lower_triangular_kron,
    solve_lower_triangular_masked_kron,
    solve_upper_triangular_kron,
    solve_upper_triangular_masked_kron,
)
from .utils import (
    flatten_cholesky_unique,
    unflatten_cholesky_unique,
    xx_t,
)

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
]

_DT = torch.float64


def _param(value):
    t = torch.as_tensor(value, dtype=_DT)
    return t.clone().detach().requires_grad_(True)


class CovBase(abc.ABC):
    """Base class for residual covariances (logdet / solve / params).

    ``to(device)`` moves every tensor attribute (templates, optimizable
    parameters) so the whole likelihood graph runs on that device —
    call it BEFORE handing parameters to an optimizer.
    """

    def __init__(self, size):
        self.size = size
        self._device = torch.device("cpu")

    def to(self, device):
        device = torch.device(device)

        def _move(val):
            moved = val.detach().to(device)
            moved.requires_grad_(val.requires_grad)
            return moved

        for name, val in list(self.__dict__.items()):
            if torch.is_tensor(val):
                setattr(self, name, _move(val))
            elif isinstance(val, list) and val and \
                    all(torch.is_tensor(v) for v in val):
                setattr(self, name, [_move(v) for v in val])
        self._device = device
        return self

    def _zero(self):
        return torch.zeros((), dtype=_DT, device=self._device)

    def _eye(self):
        return torch.eye(self.size, dtype=_DT, device=self._device)

    @property
    def logp(self):
        """Regularization log-prob (0 unless a subclass adds a prior)."""
        return self._zero()

    @abc.abstractmethod
    def get_optimize_vars(self):
        """Torch parameter tensors to optimize for this covariance."""

    @property
    def logdet(self):
        raise NotImplementedError

    @abc.abstractmethod
    def solve(self, X):
        """Σ⁻¹ X."""

    @property
    def _prec(self):
        return self.solve(self._eye())

    @property
    def _cov(self):
        return torch.linalg.inv(self._prec)


class CovIdentity(CovBase):
    """Identity covariance."""

    @property
    def logdet(self):
        return self._zero()

    def get_optimize_vars(self):
        return []

    def solve(self, X):
        return X

    @property
    def _prec(self):
        return self._eye()

    @property
    def _cov(self):
        return self._eye()


class CovAR1(CovBase):
    """AR(1) covariance via the BRSA-style precision template
    (I − ρD + ρ²F)/σ²; supports multiple scan runs via scan_onsets."""

    def __init__(self, size, rho=None, sigma=None, scan_onsets=None):
        super().__init__(size)
        if scan_onsets is None:
            self.run_sizes = [size]
            off = scipy.linalg.toeplitz(np.r_[0, 1, np.zeros(size - 2)])
            dia = np.diag(np.r_[0, np.ones(size - 2), 0])
        else:
            self.run_sizes = list(np.ediff1d(np.r_[scan_onsets, size]))
            off = scipy.sparse.block_diag(
                [scipy.linalg.toeplitz(np.r_[0, 1, np.zeros(r - 2)])
                 for r in self.run_sizes]).toarray()
            dia = scipy.sparse.block_diag(
                [np.diag(np.r_[0, np.ones(r - 2), 0])
                 for r in self.run_sizes]).toarray()
        self.offdiag_template = torch.as_tensor(off, dtype=_DT)
        self.diag_template = torch.as_tensor(dia, dtype=_DT)
        self._identity = torch.eye(size, dtype=_DT)

        self.log_sigma = _param(torch.randn(1, dtype=_DT)
                                if sigma is None else [np.log(sigma)])
        self.rho_unc = _param(
            torch.randn(1, dtype=_DT) if rho is None
            else [scipy.special.logit(rho / 2 + 0.5)])

    @property
    def logdet(self):
        rho = 2 * torch.sigmoid(self.rho_unc) - 1
        run_sizes = torch.as_tensor(self.run_sizes, dtype=_DT,
                                    device=self._device)
        return torch.sum(2 * run_sizes * self.log_sigma
                         - torch.log(1 - rho ** 2))

    @property
    def _prec(self):
        rho = 2 * torch.sigmoid(self.rho_unc) - 1
        sigma = torch.exp(self.log_sigma)
        return (self._identity - rho * self.offdiag_template
                + rho ** 2 * self.diag_template) / sigma ** 2

    def get_optimize_vars(self):
        return [self.rho_unc, self.log_sigma]

    def solve(self, X):
        return self._prec @ X


class CovIsotropic(CovBase):
    """Scaled identity."""

    def __init__(self, size, var=None):
        super().__init__(size)
        self.log_var = _param(torch.randn(1, dtype=_DT)
                              if var is None else [np.log(var)])

    @property
    def var(self):
        return torch.exp(self.log_var)

    @property
    def logdet(self):
        return (self.size * self.log_var).sum()

    def get_optimize_vars(self):
        return [self.log_var]

    def solve(self, X):
        return X / self.var


class CovDiagonal(CovBase):
    """Independent per-element variances (parameterized as log-precision)."""

    def __init__(self, size, diag_var=None):
        super().__init__(size)
        self.logprec = _param(torch.randn(size, dtype=_DT)
                              if diag_var is None
                              else np.log(1 / np.asarray(diag_var)))

    @property
    def logdet(self):
        return -torch.sum(self.logprec)

    def get_optimize_vars(self):
        return [self.logprec]

    def solve(self, X):
        return torch.exp(self.logprec)[:, None] * X


class CovDiagonalGammaPrior(CovDiagonal):
    """Diagonal covariance with an inverse-gamma prior on precisions."""

    def __init__(self, size, sigma=None, alpha=1.5, beta=1e-10):
        super().__init__(size, sigma)
        self.alpha = torch.as_tensor(alpha, dtype=_DT)
        self.beta = torch.as_tensor(beta, dtype=_DT)

    @property
    def logp(self):
        ig = torch.distributions.InverseGamma(self.alpha, self.beta)
        return torch.sum(ig.log_prob(torch.exp(self.logprec)))


class CovUnconstrainedCholesky(CovBase):
    """Unconstrained covariance Σ = LLᵀ via a flattened Cholesky."""

    def __init__(self, size=None, Sigma=None):
        if (size is None) == (Sigma is None):
            raise RuntimeError("Must pass either Sigma or size but not both")
        if Sigma is not None:
            size = Sigma.shape[0]
        super().__init__(size)
        npar = (size * (size + 1)) // 2
        if Sigma is None:
            self.L_flat = _param(torch.randn(npar, dtype=_DT))
        else:
            self.L_flat = _param(flatten_cholesky_unique(
                np.linalg.cholesky(Sigma)))

    @property
    def L(self):
        return unflatten_cholesky_unique(self.L_flat)

    @property
    def logdet(self):
        return 2 * torch.sum(torch.log(torch.diagonal(self.L)))

    def get_optimize_vars(self):
        return [self.L_flat]

    def solve(self, X):
        return torch.cholesky_solve(X, self.L)


class CovUnconstrainedCholeskyWishartReg(CovUnconstrainedCholesky):
    """Cholesky-parameterized covariance with a weak Wishart prior
    (Chung et al. 2015) pushing it away from singularity."""

    def __init__(self, size, Sigma=None):
        super().__init__(size)
        self._scale_tril = 1e5 * torch.eye(size, dtype=_DT)
        self._df = torch.as_tensor(float(size + 2), dtype=_DT)

    @property
    def logp(self):
        wishart = torch.distributions.Wishart(
            df=self._df, scale_tril=self._scale_tril)
        return wishart.log_prob(xx_t(self.L))


class CovUnconstrainedInvCholesky(CovBase):
    """Unconstrained covariance parameterized by its PRECISION Cholesky
    (saves a solve per optimization step)."""

    def __init__(self, size=None, invSigma=None):
        if (size is None) == (invSigma is None):
            raise RuntimeError(
                "Must pass either invSigma or size but not both")
        if invSigma is not None:
            size = invSigma.shape[0]
        super().__init__(size)
        npar = (size * (size + 1)) // 2
        if invSigma is None:
            self.Linv_flat = _param(torch.randn(npar, dtype=_DT))
        else:
            self.Linv_flat = _param(flatten_cholesky_unique(
                np.linalg.cholesky(invSigma)))

    @property
    def Linv(self):
        return unflatten_cholesky_unique(self.Linv_flat)

    @property
    def logdet(self):
        return -2 * torch.sum(torch.log(torch.diagonal(self.Linv)))

    def get_optimize_vars(self):
        return [self.Linv_flat]

    def solve(self, X):
        # Σ⁻¹ = Linv Linvᵀ (consistent with Cholesky init from invSigma;
        # the reference's LᵀL form is equivalent for the free
        # parameterization but wrong for a supplied invSigma)
        Linv = self.Linv
        return Linv @ (Linv.T @ X)


class CovKroneckerFactored(CovBase):
    """Kronecker-product covariance Σ = ⊗_i L_i L_iᵀ (optionally masked)."""

    def __init__(self, sizes, Sigmas=None, mask=None):
        if not isinstance(sizes, list):
            raise TypeError("sizes is not a list")
        self._device = torch.device("cpu")
        self.sizes = sizes
        self.nfactors = len(sizes)
        self.size = int(np.prod(np.array(sizes), dtype=np.int64))
        npar = [(s * (s + 1)) // 2 for s in self.sizes]
        if Sigmas is None:
            self.Lflat = [_param(torch.randn(npar[i], dtype=_DT))
                          for i in range(self.nfactors)]
        else:
            self.Lflat = [_param(flatten_cholesky_unique(
                np.linalg.cholesky(Sigmas[i])))
                for i in range(self.nfactors)]
        self.mask = (None if mask is None
                     else torch.as_tensor(mask))
        if self.mask is not None:
            self.size = int(self.mask.sum())

    @property
    def L(self):
        return [unflatten_cholesky_unique(f) for f in self.Lflat]

    def get_optimize_vars(self):
        return self.Lflat

    @property
    def logdet(self):
        if self.mask is None:
            n_list = torch.as_tensor([m.shape[0] for m in self.L],
                                     dtype=_DT, device=self._device)
            n_prod = torch.prod(n_list)
            logdets = torch.stack([
                torch.sum(torch.log(torch.diagonal(m))) for m in self.L])
            return 2.0 * torch.sum(logdets * n_prod / n_list)
        n_list = [m.shape[0] for m in self.L]
        mask_reshaped = self.mask.reshape(n_list)
        logdet = self._zero()
        for i in range(self.nfactors):
            dims = [d for d in range(self.nfactors) if d != i]
            counts = mask_reshaped.sum(dim=dims).to(_DT)
            logdet = logdet + torch.sum(
                torch.log(torch.diagonal(self.L[i])) * counts)
        return 2.0 * logdet

    def solve(self, X):
        if self.mask is None:
            z = solve_lower_triangular_kron(self.L, X)
            return solve_upper_triangular_kron(self.L, z)
        z = solve_lower_triangular_masked_kron(self.L, X, self.mask)
        return solve_upper_triangular_masked_kron(self.L, z, self.mask)
