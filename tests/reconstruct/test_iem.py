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
