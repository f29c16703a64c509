"""Matnormal: covariances vs dense scipy reference densities, solvers,
regression and MNRSA recovery (mirrors the reference's test strategy of
checking TF likelihoods against scipy)."""

import numpy as np
import pytest
import torch
from scipy.stats import multivariate_normal, wishart

from brainiak_amd.matnormal import (
    MNRSA,
    CovAR1,
    CovDiagonal,
    CovIdentity,
    CovIsotropic,
    CovKroneckerFactored,
    CovUnconstrainedCholesky,
    CovUnconstrainedInvCholesky,
    MatnormalRegression,
    matnorm_logp,
    matnorm_logp_marginal_row,
)
from brainiak_amd.matnormal.utils import (
    flatten_cholesky_unique,
    rmn,
    unflatten_cholesky_unique,
)

DT = torch.float64


def _dense_matnorm_logp(x, row_cov, col_cov):
    """scipy reference: vec(X) ~ N(0, colcov ⊗ rowcov)."""
    full = np.kron(col_cov, row_cov)
    return multivariate_normal.logpdf(x.T.ravel(), mean=None, cov=full)


def _spd(rng, n):
    a = rng.randn(n, n)
    return a @ a.T + n * np.eye(n)


def test_cholesky_flatten_roundtrip(seeded_rng):
    S = _spd(seeded_rng, 5)
    L = np.linalg.cholesky(S)
    flat = flatten_cholesky_unique(L)
    back = unflatten_cholesky_unique(flat)
    assert np.allclose(back.numpy(), L, atol=1e-12)


@pytest.mark.parametrize("cov_factory", [
    lambda n, rng: (CovIdentity(n), np.eye(n)),
    lambda n, rng: (CovIsotropic(n, var=2.5), 2.5 * np.eye(n)),
    lambda n, rng: (CovDiagonal(n, diag_var=np.arange(1, n + 1).astype(
        float)), np.diag(np.arange(1, n + 1).astype(float))),
    lambda n, rng: (lambda S: (CovUnconstrainedCholesky(Sigma=S), S))(
        _spd(rng, n)),
    lambda n, rng: (lambda S: (CovUnconstrainedInvCholesky(
        invSigma=np.linalg.inv(S)), S))(_spd(rng, n)),
])
def test_cov_logdet_and_solve(cov_factory, seeded_rng):
    n = 6
    cov, dense = cov_factory(n, seeded_rng)
    assert np.isclose(float(cov.logdet), np.linalg.slogdet(dense)[1],
                      atol=1e-6)
    X = torch.as_tensor(seeded_rng.randn(n, 3), dtype=DT)
    expected = np.linalg.solve(dense, X.numpy())
    assert np.allclose(cov.solve(X).detach().numpy(), expected, atol=1e-8)


def test_cov_ar1_matches_analytic(seeded_rng):
    n = 20
    rho, sigma = 0.4, 1.3
    cov = CovAR1(n, rho=rho, sigma=sigma)
    prec = cov._prec.detach().numpy()
    # BRSA-style stationary precision: (I - rho*D + rho^2*F)/sigma^2
    D = np.zeros((n, n))
    for i in range(n - 1):
        D[i, i + 1] = D[i + 1, i] = 1
    F = np.diag(np.r_[0, np.ones(n - 2), 0])
    expected = (np.eye(n) - rho * D + rho ** 2 * F) / sigma ** 2
    assert np.allclose(prec, expected, atol=1e-8)
    # logdet consistent with |prec|
    assert np.isclose(float(cov.logdet),
                      -np.linalg.slogdet(expected)[1], atol=1e-6)


def test_cov_kron_solve_and_logdet(seeded_rng):
    s1 = _spd(seeded_rng, 3)
    s2 = _spd(seeded_rng, 4)
    cov = CovKroneckerFactored([3, 4], Sigmas=[s1, s2])
    dense = np.kron(s1, s2)
    assert np.isclose(float(cov.logdet), np.linalg.slogdet(dense)[1],
                      atol=1e-6)
    X = torch.as_tensor(seeded_rng.randn(12, 2), dtype=DT)
    expected = np.linalg.solve(dense, X.numpy())
    assert np.allclose(cov.solve(X).detach().numpy(), expected, atol=1e-6)


def test_matnorm_logp_vs_scipy(seeded_rng):
    rows, cols = 5, 4
    row_S = _spd(seeded_rng, rows)
    col_S = _spd(seeded_rng, cols)
    x_np = rmn(row_S, col_S)
    x = torch.as_tensor(x_np, dtype=DT)
    lp = matnorm_logp(x, CovUnconstrainedCholesky(Sigma=row_S),
                      CovUnconstrainedCholesky(Sigma=col_S))
    ref = _dense_matnorm_logp(x_np, row_S, col_S)
    assert np.isclose(float(lp), ref, atol=1e-6)


def test_matnorm_logp_marginal_row_vs_dense(seeded_rng):
    rows, cols, k = 6, 3, 2
    row_S = _spd(seeded_rng, rows)
    col_S = _spd(seeded_rng, cols)
    Q = _spd(seeded_rng, k)
    A = seeded_rng.randn(rows, k)
    x_np = seeded_rng.randn(rows, cols)
    x = torch.as_tensor(x_np, dtype=DT)
    lp = matnorm_logp_marginal_row(
        x, CovUnconstrainedCholesky(Sigma=row_S),
        CovUnconstrainedCholesky(Sigma=col_S),
        torch.as_tensor(A, dtype=DT), CovUnconstrainedCholesky(Sigma=Q))
    ref = _dense_matnorm_logp(x_np, row_S + A @ Q @ A.T, col_S)
    assert np.isclose(float(lp), ref, atol=1e-6)


def test_matnormal_regression_recovers_beta(seeded_rng):
    T, V, C = 60, 8, 3
    X = seeded_rng.randn(T, C)
    beta_true = seeded_rng.randn(C, V) * 2
    Y = X @ beta_true + 0.1 * seeded_rng.randn(T, V)
    model = MatnormalRegression(time_cov=CovIdentity(T),
                                space_cov=CovIsotropic(V))
    model.fit(X, Y)
    assert np.allclose(model.beta_, beta_true, atol=0.2)
    pred = model.predict(X)
    assert np.corrcoef(pred.ravel(), Y.ravel())[0, 1] > 0.95
    X_dec = model.calibrate(Y)
    assert np.corrcoef(X_dec.ravel(), X.ravel())[0, 1] > 0.9


def test_mnrsa_recovers_planted_covariance(seeded_rng):
    """Generative setup mirrors the reference's MNRSA oracle
    (ref tests/matnormal/test_matnormal_rsa.py): block-structured U,
    beta ~ MN(U, I), nuisance regressors, diagonal temporal noise."""
    from brainiak_amd.matnormal import CovDiagonal
    from brainiak_amd.utils.utils import cov2corr
    n_C, n_T, n_V, n_nureg = 8, 120, 120, 3
    U = np.eye(n_C) * 0.6
    U[4:6, 4:6] = 0.8
    for cond in range(4, 6):
        U[cond, cond] = 1

    beta = rmn(U, np.eye(n_V))
    X = seeded_rng.randn(n_T, n_C)
    beta_0 = rmn(np.eye(n_nureg), np.eye(n_V))
    X_0 = seeded_rng.randn(n_T, n_nureg)
    timecov_true = np.diag(np.abs(seeded_rng.randn(n_T)))
    Y = X @ beta + X_0 @ beta_0 + rmn(timecov_true, np.eye(n_V))

    model = MNRSA(time_cov=CovDiagonal(n_T),
                  space_cov=CovIdentity(n_V), n_nureg=n_nureg)
    model.fit(Y, X, naive_init=True)
    rmse = np.mean((model.C_ - cov2corr(U)) ** 2) ** 0.5
    assert rmse < 0.15


def test_wishart_reg_logp(seeded_rng):
    from brainiak_amd.matnormal import CovUnconstrainedCholeskyWishartReg
    cov = CovUnconstrainedCholeskyWishartReg(4)
    lp = float(cov.logp)
    # matches scipy wishart on the same Sigma
    L = cov.L.detach().numpy()
    S = L @ L.T
    ref = wishart.logpdf(S, df=6, scale=1e10 * np.eye(4))
    assert np.isclose(lp, ref, rtol=1e-5, atol=1e-3)


def test_kron_mult_matches_dense(seeded_rng):
    import torch
    from brainiak_amd.utils.kronecker_solvers import kron_mult
    A = torch.as_tensor(seeded_rng.randn(3, 3))
    B = torch.as_tensor(seeded_rng.randn(4, 4))
    X = torch.as_tensor(seeded_rng.randn(12, 5))
    ref = torch.kron(A, B) @ X
    assert torch.allclose(kron_mult([A, B], X), ref, atol=1e-10)
