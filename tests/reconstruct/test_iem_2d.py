"""InvertedEncoding2D oracle breadth (ports the reference's 2-D test
behaviors, ref tests/reconstruct/test_iem.py:85-345)."""

import numpy as np
import pytest

from brainiak_amd.reconstruct.iem import (
    InvertedEncoding1D,
    InvertedEncoding2D,
)


def test_instantiate_improper_range():
    with pytest.raises(ValueError):
        InvertedEncoding1D(6, 5, 'halfcircular', range_start=20,
                           range_stop=0)
    with pytest.raises(ValueError):
        InvertedEncoding2D(stim_xlim=[0, -1], stim_ylim=[0, -1],
                           stimulus_resolution=[10, 10])
    with pytest.raises(ValueError):
        InvertedEncoding2D(stim_xlim=[0], stim_ylim=[-1, 0],
                           stimulus_resolution=10)


def test_2d_data_amount_and_dimensions(seeded_rng):
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-1, 1],
                           stimulus_resolution=10, stim_radius=0.5)
    m.define_basis_functions_sqgrid(nchannels=4)
    with pytest.raises(ValueError):     # fewer trials than channels
        m.fit(np.random.rand(5, 100), np.random.rand(5, 2))
    with pytest.raises(ValueError):     # 3-D data
        m.fit(np.random.rand(30, 10, 2), np.random.rand(30, 2))


def test_2d_stimulus_resolution():
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-1, 1],
                           stimulus_resolution=10)
    assert len(m.stim_pixels[0]) == 10
    assert len(m.stim_pixels[1]) == 10
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-2, 2],
                           stimulus_resolution=[10, 20])
    assert len(m.stim_pixels[0]) == 10
    assert len(m.stim_pixels[1]) == 20


def test_2d_custom_channels_and_inconsistent(seeded_rng):
    nchan, res = 8, 10
    channels = np.random.rand(nchan, res * res) * 2 - 1
    bds = [-1, 1]
    m = InvertedEncoding2D(stim_xlim=bds, stim_ylim=bds,
                           stimulus_resolution=res, chan_xlim=bds,
                           chan_ylim=bds, channels=channels)
    assert m.n_channels == nchan
    with pytest.raises(ValueError):     # channels over wrong pixel count
        InvertedEncoding2D(stim_xlim=bds, stim_ylim=bds,
                           stimulus_resolution=10,
                           channels=np.random.rand(5, 5))
    with pytest.raises(ValueError):     # inconsistent property change
        m.set_params(n_channels=nchan - 1)


def test_get_2d_params():
    bds = [-1, 1]
    m = InvertedEncoding2D(stim_xlim=bds, stim_ylim=bds,
                           stimulus_resolution=10)
    params = m.get_params()
    assert np.all(params.get('stim_fov')[0] == bds)
    assert params.get('xp').size == 100


def test_2d_cosine_shape_and_mask(seeded_rng):
    nchan, res = 8, 10
    bds = [-1, 1]
    m = InvertedEncoding2D(stim_xlim=bds, stim_ylim=bds,
                           stimulus_resolution=res,
                           channels=np.random.rand(nchan, res * res))
    sz = m._2d_cosine_fwhm_to_sz(1)
    fcn = m._make_2d_cosine(m.xp.reshape(-1, 1), m.yp.reshape(-1, 1),
                            np.linspace(bds[0], bds[1], nchan),
                            np.linspace(bds[0], bds[1], nchan), sz)
    assert fcn.shape == (nchan, res * res)
    xd = np.diff(m.stim_pixels[0])[0]
    nval = np.nonzero(fcn[0, :])[0].size
    assert nval * (xd ** 2) <= sz ** 2


def test_2d_cos_size_roundtrip():
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-1, 1],
                           stimulus_resolution=10)
    s = np.random.rand()
    fwhm = m._2d_cosine_sz_to_fwhm(s)
    assert np.isclose(s, m._2d_cosine_fwhm_to_sz(fwhm))
    assert np.isclose(fwhm,
                      m._2d_cosine_sz_to_fwhm(
                          m._2d_cosine_fwhm_to_sz(fwhm)))


def test_square_basis_grid():
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-1, 1],
                           stimulus_resolution=10)
    _, centers = m.define_basis_functions_sqgrid(nchannels=8)
    assert centers.shape[0] == 64
    xsp = np.round(np.diff(centers[:, 0]), 5)
    assert xsp[0] == xsp[28] == xsp[-1]


def test_triangular_basis_grid():
    grid_rad = 3
    m = InvertedEncoding2D(stim_xlim=[-1, 1], stim_ylim=[-1, 1],
                           stimulus_resolution=10)
    _, centers = m.define_basis_functions_trigrid(grid_rad)
    assert centers.shape[0] == (grid_rad * 2 + 1) * (grid_rad * 2)
    # odd rows offset half a step; x spacings uniform within rows
    xsp = np.round(np.diff(centers[:, 0]), 4)
    assert xsp[0] == xsp[-1]
    ysp = np.diff(centers[:, 1])
    pos = ysp[ysp > 0]
    x_dist = 2.0 / (grid_rad * 2)
    assert np.allclose(pos, x_dist * np.sqrt(3) * 0.5)


def _grid_design(nobs, xlim, ylim):
    sxx, syy = np.meshgrid(np.linspace(xlim[0], xlim[1], 10),
                           np.linspace(ylim[0], ylim[1], 10))
    return np.hstack((sxx.reshape(-1, 1), syy.reshape(-1, 1)))


def test_fit_requires_some_radius(seeded_rng):
    xlim, ylim = [-6, 6], [-3, 3]
    yd = _grid_design(100, xlim, ylim)
    Xd = np.random.rand(100, 50)
    m = InvertedEncoding2D(stim_xlim=xlim, stim_ylim=ylim,
                           stimulus_resolution=[50, 50],
                           stim_radius=None)
    m.define_basis_functions_sqgrid(nchannels=[6, 4])
    with pytest.raises(ValueError):
        m.fit(Xd, yd)


def test_fit_2d_radius_list_and_custom_C(seeded_rng):
    xlim, ylim = [-6, 6], [-3, 3]
    yd = _grid_design(100, xlim, ylim)
    Xd = np.random.rand(100, 50)
    m = InvertedEncoding2D(stim_xlim=xlim, stim_ylim=ylim,
                           stimulus_resolution=[50, 50],
                           stim_radius=np.random.rand(100) + 0.5)
    m.define_basis_functions_sqgrid(nchannels=[6, 4])
    m.fit(Xd, yd)
    assert m.W_.shape[0] == 50
    # explicit channel-activation design bypasses stimulus handling
    m2 = InvertedEncoding2D(stim_xlim=xlim, stim_ylim=ylim,
                            stimulus_resolution=[50, 50])
    m2.define_basis_functions_sqgrid(nchannels=[6, 4])
    C = np.random.rand(100, m2.n_channels)
    m2.fit(Xd, yd, C=C)
    pred = m2.predict(np.random.rand(5, 50))
    assert pred.shape == (5, 2)


def test_2d_recovery_and_scores(seeded_rng):
    """Planted 2-D stimulus positions are recovered within the grid
    spacing, and both scoring surfaces behave."""
    rng = seeded_rng
    xlim, ylim, res = [-2, 2], [-2, 2], [40, 40]
    m = InvertedEncoding2D(stim_xlim=xlim, stim_ylim=ylim,
                           stimulus_resolution=res, stim_radius=0.5,
                           channel_exp=5)
    m.define_basis_functions_sqgrid(nchannels=6)
    centers = np.array([[x, y] for x in np.linspace(-1.5, 1.5, 6)
                        for y in np.linspace(-1.5, 1.5, 6)])
    C = m._define_trial_activations(centers)
    nvox = 80
    W = rng.rand(nvox, m.n_channels)
    X = C @ W.T + 0.05 * rng.randn(centers.shape[0], nvox)
    m.fit(X, centers)
    pred = m.predict(X)
    err = np.linalg.norm(pred - centers, axis=1)
    assert np.median(err) < 0.6, np.median(err)
    r2 = m.score(X, centers)
    assert r2.shape == (centers.shape[0],)
    d = m.score_against_reconstructed(X[:1], m.predict_feature_responses(
        X[:1]), metric="euclidean")
    assert np.all(np.isfinite(d))
    with pytest.raises(ValueError):
        m.score_against_reconstructed(X[:1], X[:1], metric="manhattan")
