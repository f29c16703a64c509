import numpy as np
import pytest

from brainiak_amd.reconstruct.iem import (
    InvertedEncoding1D,
    InvertedEncoding2D,
)


def _iem1d_data(rng, n_trials=60, n_voxels=30, n_channels=6):
    """Voxel data generated from a ground-truth channel model."""
    iem = InvertedEncoding1D(n_channels=n_channels)
    channels, _ = iem._define_channels()
    features = rng.choice(np.arange(0, 180, 12), size=n_trials)
    C = iem.__class__(n_channels=n_channels)
    C.channels_ = channels
    acts = C._define_trial_activations(features)   # [trials, channels]
    W = rng.rand(n_voxels, n_channels)
    X = acts @ W.T + 0.05 * rng.randn(n_trials, n_voxels)
    return X, features


def test_iem1d_fit_predict_score(seeded_rng):
    X, y = _iem1d_data(seeded_rng)
    iem = InvertedEncoding1D(n_channels=6)
    iem.fit(X, y)
    pred = iem.predict(X)
    assert pred.shape == y.shape
    # predictions close to true (circular distance in degrees)
    err = np.abs(((pred - y) + 90) % 180 - 90)
    assert np.median(err) < 20
    assert iem.score(X, y) > 0.5


def test_iem1d_validation(seeded_rng):
    with pytest.raises(ValueError):
        InvertedEncoding1D(range_start=100, range_stop=50)
    with pytest.raises(ValueError):
        InvertedEncoding1D(stimulus_mode='halfcircular', range_stop=90.)
    with pytest.raises(ValueError):
        InvertedEncoding1D(stimulus_mode='circular', range_stop=180.)
    with pytest.raises(ValueError):
        InvertedEncoding1D(n_channels=1)
    with pytest.raises(ValueError):
        InvertedEncoding1D(stimulus_mode='bogus',
                           range_start=0, range_stop=180)
    X, y = _iem1d_data(seeded_rng)
    iem = InvertedEncoding1D()
    with pytest.raises(ValueError):
        iem.fit(X[:3], y[:3])   # fewer trials than channels
    with pytest.raises(ValueError):
        iem.fit(X, y[:-2])


def test_iem1d_circular_mode(seeded_rng):
    iem = InvertedEncoding1D(n_channels=6, stimulus_mode='circular',
                             range_start=0., range_stop=360.,
                             channel_density=360)
    channels, centers = iem._define_channels()
    assert channels.shape == (6, 360)
    assert len(centers) == 6
    assert np.all(channels >= 0)


def test_iem2d_basic(seeded_rng):
    iem = InvertedEncoding2D(stim_xlim=[0, 10], stim_ylim=[0, 10],
                             stimulus_resolution=20, stim_radius=2.0,
                             channel_exp=7)
    channels, centers = iem.define_basis_functions_sqgrid(4)
    assert channels.shape[0] == 16
    # generate data from the channel model
    n_trials = 40
    stim_centers = seeded_rng.rand(n_trials, 2) * 10
    C = iem._define_trial_activations(stim_centers)
    W_true = seeded_rng.rand(25, 16)
    X = C @ W_true.T + 0.05 * seeded_rng.randn(n_trials, 25)
    iem.fit(X, stim_centers)
    pred = iem.predict(X)
    assert pred.shape == (n_trials, 2)
    err = np.linalg.norm(pred - stim_centers, axis=1)
    assert np.median(err) < 4.0


def test_iem2d_trigrid(seeded_rng):
    iem = InvertedEncoding2D(stim_xlim=[-5, 5], stim_ylim=[-5, 5],
                             stimulus_resolution=16, stim_radius=1.5)
    channels, centers = iem.define_basis_functions_trigrid(3)
    assert channels.shape[0] == centers.shape[0]
    assert channels.shape[1] == 256


def test_iem2d_validation():
    with pytest.raises(ValueError):
        InvertedEncoding2D(stim_xlim=[5, 0], stim_ylim=[0, 5],
                           stimulus_resolution=10)
    with pytest.raises(ValueError):
        InvertedEncoding2D(stim_xlim=[0], stim_ylim=[0, 5],
                           stimulus_resolution=10)


# -- round-2 depth (ref tests/reconstruct/test_iem.py:1-544) -----------------

def test_channel_basis_properties():
    iem = InvertedEncoding1D(n_channels=6)
    channels, centers = iem._define_channels()
    assert channels.shape == (6, 180)
    # centers evenly spaced over the half-circle
    assert np.allclose(np.diff(centers), np.deg2rad(30))
    # each channel peaks at (approximately) its own center
    for k, c in enumerate(np.rad2deg(centers)):
        peak = iem.channel_domain[np.argmax(channels[k])]
        d = abs(((peak - c) + 90) % 180 - 90)
        assert d <= 1.0, (k, peak, c)
    # non-negative half-rectified basis
    assert channels.min() >= 0
    # channels sum to an (almost) flat population response
    pop = channels.sum(axis=0)
    assert pop.std() / pop.mean() < 0.05


def test_trial_activations_peak_at_stimulus():
    iem = InvertedEncoding1D(n_channels=6)
    iem.channels_, _ = iem._define_channels()
    stims = np.array([0.0, 30.0, 90.0, 150.0])
    C = iem._define_trial_activations(stims)
    assert C.shape == (4, 6)
    # the strongest channel for stimulus 30 deg is the channel
    # centered at 30 deg (index 1)
    assert np.argmax(C[1]) == 1
    assert np.argmax(C[2]) == 3       # 90 deg → channel 3


def test_iem1d_scoring_perfect_and_shuffled(seeded_rng):
    X, y = _iem1d_data(seeded_rng, n_trials=80)
    iem = InvertedEncoding1D(n_channels=6).fit(X, y)
    s_good = iem.score(X, y)
    s_bad = iem.score(X, seeded_rng.permutation(y))
    assert s_good > 0.5
    assert s_good > s_bad


def test_iem1d_circular_predicts(seeded_rng):
    iem = InvertedEncoding1D(n_channels=8, stimulus_mode='circular',
                             range_start=0., range_stop=360.,
                             channel_density=360)
    channels, _ = iem._define_channels()
    feats = seeded_rng.choice(np.arange(0, 360, 10), size=80)
    helper = InvertedEncoding1D(n_channels=8, stimulus_mode='circular',
                                range_start=0., range_stop=360.,
                                channel_density=360)
    helper.channels_ = channels
    acts = helper._define_trial_activations(feats)
    W = seeded_rng.rand(40, 8)
    X = acts @ W.T + 0.05 * seeded_rng.randn(80, 40)
    iem.fit(X, feats)
    pred = iem.predict(X)
    err = np.abs(((pred - feats) + 180) % 360 - 180)
    assert np.median(err) < 25
    assert iem.score(X, feats) > 0.4


def test_iem1d_get_set_params_roundtrip():
    iem = InvertedEncoding1D(n_channels=5, channel_exp=4)
    p = iem.get_params()
    assert p["n_channels"] == 5 and p["channel_exp"] == 4
    iem.set_params(n_channels=7)
    assert iem.n_channels == 7
    with pytest.raises(ValueError):
        iem.set_params(range_stop=90.)   # breaks halfcircular span


def test_iem2d_fwhm_conversion_inverts():
    iem = InvertedEncoding2D(stim_xlim=[0, 10], stim_ylim=[0, 10],
                             stimulus_resolution=10, stim_radius=2.0)
    for fwhm in (1.0, 2.5, 4.0):
        sz = iem._2d_cosine_fwhm_to_sz(fwhm)
        assert np.isclose(iem._2d_cosine_sz_to_fwhm(sz), fwhm)


def test_iem2d_cosine_bumps_properties():
    iem = InvertedEncoding2D(stim_xlim=[0, 4], stim_ylim=[0, 4],
                             stimulus_resolution=9, stim_radius=1.0)
    x = np.linspace(0, 4, 9)
    xx, yy = np.meshgrid(x, x)
    bumps = iem._make_2d_cosine(xx, yy, np.array([2.0]),
                                np.array([2.0]), 1.5)
    assert bumps.shape == (1, 81)
    grid = bumps.reshape(9, 9)
    # peak at the center, zero beyond the radius
    assert grid[4, 4] == grid.max() > 0.9
    assert grid[0, 0] == 0.0


def test_iem2d_score_against_reconstructed(seeded_rng):
    iem = InvertedEncoding2D(stim_xlim=[0, 10], stim_ylim=[0, 10],
                             stimulus_resolution=12, stim_radius=2.0)
    iem.define_basis_functions_sqgrid(3)
    n = 30
    centers = seeded_rng.rand(n, 2) * 10
    C = iem._define_trial_activations(centers)
    W = seeded_rng.rand(20, 9)
    X = C @ W.T + 0.05 * seeded_rng.randn(n, 20)
    iem.fit(X, centers)
    maps = iem.predict_feature_responses(X)
    assert maps.shape == (144, n)
    d_euc = iem.score_against_reconstructed(X, maps)
    assert d_euc.shape == (n,)
    assert np.isclose(d_euc[0], 0.0, atol=1e-5)
    d_cos = iem.score_against_reconstructed(X, maps, metric="cosine")
    assert np.all(d_cos >= -1e-9)
    with pytest.raises(ValueError):
        iem.score_against_reconstructed(X, maps, metric="bogus")


def test_iem2d_custom_channels_validation(seeded_rng):
    ch = np.abs(seeded_rng.rand(4, 100))
    iem = InvertedEncoding2D(stim_xlim=[0, 10], stim_ylim=[0, 10],
                             stimulus_resolution=10, stim_radius=2.0,
                             channels=ch)
    assert iem.n_channels == 4
    with pytest.raises(ValueError):
        InvertedEncoding2D(stim_xlim=[0, 10], stim_ylim=[0, 10],
                           stimulus_resolution=7, stim_radius=2.0,
                           channels=ch)   # 49 pixels != 100
