"""Matrix-normal log-likelihoods (torch).

Re-expression of the reference's TF likelihoods
(ref src/brainiak/matnormal/matnormal_likelihoods.py:27-429): the
Woodbury/matrix-determinant-lemma ``solve_det_marginal`` /
``solve_det_conditional`` plus the marginal/conditional row/column
matnorm log-densities.
"""

import logging

import torch

from .utils import scaled_I

logger = logging.getLogger(__name__)

__all__ = [
    "matnorm_logp",
    "matnorm_logp_conditional_col",
    "matnorm_logp_conditional_row",
    "matnorm_logp_marginal_col",
    "matnorm_logp_marginal_row",
    "solve_det_conditional",
    "solve_det_marginal",
]


def solve_det_marginal(x, sigma, A, Q):
    """(Σ + AQAᵀ)⁻¹x and log|Σ + AQAᵀ| via Woodbury + det lemma."""
    lemma_factor = torch.linalg.cholesky(Q._prec + A.T @ sigma.solve(A))
    logdet = (Q.logdet + sigma.logdet
              + 2 * torch.sum(torch.log(torch.diagonal(lemma_factor))))
    Atrp_Sinv = A.T @ sigma._prec
    prod_term = torch.cholesky_solve(Atrp_Sinv, lemma_factor)
    solve = sigma.solve(
        scaled_I(1.0, sigma.size, dtype=x.dtype,
                 device=x.device) - A @ prod_term) @ x
    return solve, logdet


def solve_det_conditional(x, sigma, A, Q):
    """(Σ − AQ⁻¹Aᵀ)⁻¹x and log|Σ − AQ⁻¹Aᵀ|."""
    lemma_factor = torch.linalg.cholesky(Q._cov - A.T @ sigma.solve(A))
    logdet = (-Q.logdet + sigma.logdet
              + 2 * torch.sum(torch.log(torch.diagonal(lemma_factor))))
    Atrp_Sinv = A.T @ sigma._prec
    prod_term = torch.cholesky_solve(Atrp_Sinv, lemma_factor)
    solve = sigma.solve(
        scaled_I(1.0, sigma.size, dtype=x.dtype,
                 device=x.device) + A @ prod_term) @ x
    return solve, logdet


def _mnorm_logp_internal(colsize, rowsize, logdet_row, logdet_col,
                         solve_row, solve_col):
    log2pi = 1.8378770664093453
    denominator = (-rowsize * colsize * log2pi - colsize * logdet_row
                   - rowsize * logdet_col)
    numerator = -torch.trace(solve_col @ solve_row)
    return 0.5 * (numerator + denominator)


def matnorm_logp(x, row_cov, col_cov):
    """Centered matrix-normal log density of x [rows, cols]."""
    rowsize, colsize = float(x.shape[0]), float(x.shape[1])
    solve_col = col_cov.solve(x.T)
    solve_row = row_cov.solve(x)
    return _mnorm_logp_internal(colsize, rowsize, row_cov.logdet,
                                col_cov.logdet, solve_row, solve_col)


def matnorm_logp_marginal_row(x, row_cov, col_cov, marg, marg_cov):
    """logp of Y ~ MN(0, R + AQAᵀ, C)."""
    rowsize, colsize = float(x.shape[0]), float(x.shape[1])
    solve_col = col_cov.solve(x.T)
    solve_row, logdet_row = solve_det_marginal(x, row_cov, marg, marg_cov)
    return _mnorm_logp_internal(colsize, rowsize, logdet_row,
                                col_cov.logdet, solve_row, solve_col)


def matnorm_logp_marginal_col(x, row_cov, col_cov, marg, marg_cov):
    """logp of Y ~ MN(0, R, C + AᵀQA)."""
    rowsize, colsize = float(x.shape[0]), float(x.shape[1])
    solve_row = row_cov.solve(x)
    solve_col, logdet_col = solve_det_marginal(x.T, col_cov, marg.T,
                                               marg_cov)
    return _mnorm_logp_internal(colsize, rowsize, row_cov.logdet,
                                logdet_col, solve_row, solve_col)


def matnorm_logp_conditional_row(x, row_cov, col_cov, cond, cond_cov):
    """logp with conditioned row covariance Σ_r − AQ⁻¹Aᵀ."""
    rowsize, colsize = float(x.shape[0]), float(x.shape[1])
    solve_col = col_cov.solve(x.T)
    solve_row, logdet_row = solve_det_conditional(x, row_cov, cond,
                                                  cond_cov)
    return _mnorm_logp_internal(colsize, rowsize, logdet_row,
                                col_cov.logdet, solve_row, solve_col)


def matnorm_logp_conditional_col(x, row_cov, col_cov, cond, cond_cov):
    """logp with conditioned column covariance Σ_c − AᵀQ⁻¹A."""
    rowsize, colsize = float(x.shape[0]), float(x.shape[1])
    solve_row = row_cov.solve(x)
    solve_col, logdet_col = solve_det_conditional(x.T, col_cov, cond.T,
                                                  cond_cov)
    return _mnorm_logp_internal(colsize, rowsize, row_cov.logdet,
                                logdet_col, solve_row, solve_col)
