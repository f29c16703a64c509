import numpy as np
import pytest

from brainiak_amd.factoranalysis.htfa import HTFA
from brainiak_amd.parallel import spawn_ranks


def _multi_subject_data(rng, n_subj=2, K=2, n_voxels=120, n_tr=30):
    centers = rng.rand(K, 3) * 20
    widths = np.full((K, 1), 10.0)
    X, R = [], []
    for _ in range(n_subj):
        coords = rng.rand(n_voxels, 3) * 20
        d2 = ((coords[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
        F = np.exp(-d2 / widths.ravel()[None, :])
        W = rng.randn(K, n_tr)
        X.append(F @ W + 0.1 * rng.randn(n_voxels, n_tr))
        R.append(coords)
    return X, R, centers


def test_htfa_fit_serial(seeded_rng):
    X, R, centers = _multi_subject_data(seeded_rng)
    htfa = HTFA(K=2, n_subj=2, max_global_iter=3, max_local_iter=2,
                voxel_ratio=1.0, tr_ratio=1.0, max_voxel=120, max_tr=30,
                device="cpu")
    htfa.fit(X, R)
    assert htfa.global_posterior_ is not None
    assert htfa.local_weights_.shape == (2 * 2 * 30,)
    # template centers should land near the true generating centers
    est = htfa.get_centers(htfa.global_posterior_)
    from scipy.spatial.distance import cdist
    d = cdist(centers, est)
    assert np.max(d.min(axis=1)) < 8.0


def test_htfa_input_validation(seeded_rng):
    X, R, _ = _multi_subject_data(seeded_rng)
    htfa = HTFA(K=2, n_subj=2, device="cpu")
    with pytest.raises(TypeError):
        htfa.fit(X[0], R)
    with pytest.raises(TypeError):
        htfa.fit(X, R[0])
    with pytest.raises(ValueError):
        htfa.fit([], [])


def _dist_htfa(ctx, outfile):
    rng = np.random.RandomState(9)
    X, R, _ = _multi_subject_data(rng, n_subj=2)
    mine = [i for i in range(2) if i % ctx.world_size == ctx.rank]
    htfa = HTFA(K=2, n_subj=2, max_global_iter=2, max_local_iter=2,
                voxel_ratio=1.0, tr_ratio=1.0, max_voxel=120, max_tr=30,
                comm=ctx, device="cpu")
    htfa.fit([X[i] for i in mine], [R[i] for i in mine])
    if ctx.rank == 0:
        np.save(outfile, htfa.global_posterior_)


@pytest.mark.slow
def test_htfa_distributed_runs(tmp_path):
    out = str(tmp_path / "post.npy")
    spawn_ranks(_dist_htfa, world_size=2, args=(out,))
    post = np.load(out)
    assert post.shape == (2 * (3 + 2 + 6),)  # K*(n_dim+2+cov_vec_size)
    assert np.all(np.isfinite(post))


def test_map_update_posterior_matches_per_factor_reference(seeded_rng):
    """The batched MAP update equals a straightforward per-factor
    Gaussian-posterior computation."""
    from brainiak_amd.factoranalysis.htfa import HTFA
    from brainiak_amd.utils.utils import from_sym_2_tri, from_tri_2_sym
    K, D, S = 4, 3, 5
    m = HTFA(K=K, n_subj=S, max_global_iter=1, max_local_iter=1)
    m.n_dim = D
    m.cov_vec_size = D * (D + 1) // 2
    m.prior_size = K * (D + 1)
    m.prior_bcast_size = K * (D + 2 + m.cov_vec_size)
    m.get_map_offset()
    rng = seeded_rng
    # a random-but-valid global prior
    prior = np.zeros(m.prior_bcast_size)
    prior[:K * D] = rng.randn(K * D)
    prior[K * D:K * (D + 1)] = 1 + rng.rand(K)
    covs = []
    for k in range(K):
        a = rng.randn(D, D)
        covs.append(from_sym_2_tri(a @ a.T + np.eye(D)))
    prior[int(m.map_offset[2]):int(m.map_offset[2])
          + K * m.cov_vec_size] = np.concatenate(covs)
    prior[int(m.map_offset[3]):int(m.map_offset[3]) + K] = \
        0.5 + rng.rand(K)
    m.global_prior_ = prior
    g = rng.randn(D, D)
    m.global_centers_cov_scaled = (g @ g.T + np.eye(D)) / S
    m.global_widths_var_scaled = 0.3
    m.gather_posterior = rng.randn(S * m.prior_size)
    m._map_update_posterior()

    # independent per-factor reference
    stacked = m.gather_posterior.reshape(S, m.prior_size)
    for k in range(K):
        mu_p = prior[k * D:(k + 1) * D]
        cov_p = from_tri_2_sym(covs[k], D)
        cov_p = cov_p + cov_p.T - np.diag(np.diag(cov_p))
        xbar = stacked[:, k * D:(k + 1) * D].mean(axis=0)
        inv = np.linalg.inv(cov_p + m.global_centers_cov_scaled)
        mu_star = cov_p @ inv @ xbar \
            + m.global_centers_cov_scaled @ inv @ mu_p
        assert np.allclose(
            m.global_posterior_[k * D:(k + 1) * D], mu_star, atol=1e-10)
        cov_star = cov_p @ inv @ m.global_centers_cov_scaled
        got_tri = m.global_posterior_[
            int(m.map_offset[2]) + k * m.cov_vec_size:
            int(m.map_offset[2]) + (k + 1) * m.cov_vec_size]
        assert np.allclose(got_tri, from_sym_2_tri(cov_star),
                           atol=1e-10)
        wp = prior[K * D + k]
        wv = prior[int(m.map_offset[3]) + k]
        xw = stacked[:, K * D + k].mean()
        denom = wv + m.global_widths_var_scaled
        w_star = (wv * xw + m.global_widths_var_scaled * wp) / denom
        assert np.isclose(
            m.global_posterior_[int(m.map_offset[1]) + k], w_star)
