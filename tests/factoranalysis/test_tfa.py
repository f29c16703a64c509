import numpy as np
import pytest

from brainiak_amd.factoranalysis.tfa import TFA


def _rbf_data(rng, K=3, n_voxels=200, n_tr=40, noise=0.05):
    """Synthetic data generated exactly by the TFA model."""
    coords = rng.rand(n_voxels, 3) * 20
    centers = rng.rand(K, 3) * 20
    widths = np.full((K, 1), 12.0)
    d2 = ((coords[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
    F = np.exp(-d2 / widths.ravel()[None, :])
    W = rng.randn(K, n_tr)
    X = F @ W + noise * rng.randn(n_voxels, n_tr)
    return X, coords, centers, widths


def test_tfa_fit_basic(seeded_rng):
    X, R, centers, widths = _rbf_data(seeded_rng)
    tfa = TFA(K=3, max_iter=5, max_num_voxel=200, max_num_tr=40,
              verbose=False, device="cpu")
    tfa.fit(X, R)
    assert tfa.F_.shape == (200, 3)
    assert tfa.W_.shape == (3, 40)
    # factors should be valid RBFs in (0, 1]
    assert np.all(tfa.F_ > 0) and np.all(tfa.F_ <= 1.0 + 1e-9)
    # reconstruction should capture most of the variance
    recon = tfa.F_ @ tfa.W_
    r = np.corrcoef(recon.ravel(), X.ravel())[0, 1]
    assert r > 0.7
    # recovered centers should be near true centers (match greedily)
    est = tfa.get_centers(tfa.local_posterior_)
    from scipy.spatial.distance import cdist
    d = cdist(centers, est)
    assert np.max(d.min(axis=1)) < 6.0


def test_tfa_factor_matches_formula(seeded_rng):
    X, R, _, _ = _rbf_data(seeded_rng, n_voxels=50)
    tfa = TFA(K=2, device="cpu")
    tfa.n_dim = 3
    tfa.cov_vec_size = 6
    tfa.get_map_offset()
    centers = seeded_rng.rand(2, 3) * 20
    widths = np.array([[5.0], [9.0]])
    unique_R, inds = tfa.get_unique_R(R)
    F = tfa.get_factors(unique_R, inds, centers, widths)
    d2 = ((R[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
    expected = np.exp(-d2 / widths.ravel()[None, :])
    assert np.allclose(F, expected, atol=1e-10)


def test_tfa_weights_rr_vs_ols(seeded_rng):
    X, R, _, _ = _rbf_data(seeded_rng, noise=0.0)
    tfa = TFA(K=3, weight_method='ols', device="cpu")
    tfa.n_dim = 3
    tfa.cov_vec_size = 6
    tfa.get_map_offset()
    F = seeded_rng.rand(200, 3) + 0.1
    W_ols = tfa.get_weights(X, F)
    expected = np.linalg.solve(F.T @ F, F.T @ X)
    assert np.allclose(W_ols, expected, atol=1e-8)
    tfa.weight_method = 'rr'
    W_rr = tfa.get_weights(X, F)
    beta = np.var(X)
    expected_rr = np.linalg.solve(F.T @ F + beta * np.eye(3), F.T @ X)
    assert np.allclose(W_rr, expected_rr, atol=1e-8)


def test_tfa_input_validation(seeded_rng):
    X, R, _, _ = _rbf_data(seeded_rng, n_voxels=50)
    tfa = TFA(K=2, device="cpu")
    with pytest.raises(TypeError):
        tfa.fit([1, 2, 3], R)
    with pytest.raises(TypeError):
        tfa.fit(X, R[:10])  # voxel mismatch
    with pytest.raises(ValueError):
        TFA(K=2, weight_method='bogus', device="cpu").fit(X, R)


def test_tfa_with_template_prior(seeded_rng):
    X, R, _, _ = _rbf_data(seeded_rng, n_voxels=100)
    t = TFA(K=2, max_iter=2, max_num_voxel=100, max_num_tr=40,
            device="cpu")
    t.n_dim = 3
    t.cov_vec_size = 6
    t.get_map_offset()
    template, _, _ = t.get_template(R)
    t2 = TFA(K=2, max_iter=2, max_num_voxel=100, max_num_tr=40,
             device="cpu")
    t2.fit(X, R, template_prior=template)
    assert t2.local_posterior_.shape == (2 * 4,)


def test_tfa_analytic_jacobian_matches_fd(seeded_rng):
    """The closed-form NLSS Jacobian equals central finite differences
    (the reference differentiates by FD; we compute it analytically
    from one factor evaluation)."""
    from brainiak_amd.utils.utils import from_sym_2_tri
    K, V, T = 3, 80, 15
    coords = seeded_rng.rand(V, 3) * 20
    tfa = TFA(K=K, device="cpu")
    tfa.n_dim = 3
    tfa.cov_vec_size = 6
    tfa.get_map_offset()
    tfa.sample_scaling = 0.7
    unique_R, inds = tfa.get_unique_R(coords)
    X = seeded_rng.randn(V, T)
    W = seeded_rng.randn(K, T)
    centers = seeded_rng.rand(K, 3) * 20
    widths = np.array([[6.], [9.], [12.]])
    est = np.hstack([centers.ravel(), widths.ravel()])
    tc = centers + seeded_rng.randn(K, 3)
    tcov = np.vstack([from_sym_2_tri(np.eye(3) * 3 + 0.5)
                      for _ in range(K)])
    args = (unique_R, inds, X, W, tc, tcov, widths + 1.0,
            np.abs(seeded_rng.rand(K, 1)) + 0.5, 0.9)
    J = tfa._jacobian_multivariate(est, *args)
    eps = 1e-6
    for i in range(len(est)):
        ep = est.copy()
        ep[i] += eps
        em = est.copy()
        em[i] -= eps
        col = (tfa._residual_multivariate(ep, *args)
               - tfa._residual_multivariate(em, *args)) / (2 * eps)
        assert np.abs(J[:, i] - col).max() < 2e-4, i
