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
