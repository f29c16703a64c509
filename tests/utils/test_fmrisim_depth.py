"""fmrisim depth battery (ref tests/utils/test_fmrisim.py:1-913):
signal-shape options, stimfunction timing/export round trips, HRF
properties, noise-component behaviors, spatial-smoothness estimation
and signal-change scaling."""

import numpy as np
import pytest

from brainiak_amd.utils import fmrisim as sim


# -- signal generation --------------------------------------------------------

@pytest.mark.parametrize("shape", ["cube", "sphere", "loop"])
def test_generate_signal_shapes_param(shape):
    dims = np.array([12, 12, 12])
    vol = sim.generate_signal(dimensions=dims,
                              feature_coordinates=np.array([[6, 6, 6]]),
                              feature_size=[3],
                              feature_type=[shape],
                              signal_magnitude=[1])
    assert vol.shape == tuple(dims)
    assert vol.max() == 1
    assert vol[6, 6, 6] == (0 if shape == "loop" else 1)
    # signal is local: corners stay empty
    assert vol[0, 0, 0] == 0


def test_generate_signal_multiple_features():
    dims = np.array([16, 16, 16])
    coords = np.array([[4, 4, 4], [12, 12, 12]])
    vol = sim.generate_signal(dimensions=dims,
                              feature_coordinates=coords,
                              feature_size=[2, 2],
                              feature_type=["cube", "cube"],
                              signal_magnitude=[1, 2])
    assert vol[4, 4, 4] == 1
    assert vol[12, 12, 12] == 2


# -- stimfunction + exports ---------------------------------------------------

def test_stimfunction_onset_accuracy():
    sf = sim.generate_stimfunction(onsets=[5, 20], event_durations=[3],
                                   total_time=30,
                                   temporal_resolution=10)
    assert sf.shape == (300, 1)
    assert sf[49, 0] == 0 and sf[50, 0] == 1     # onset at 5 s
    assert sf[79, 0] == 1 and sf[80, 0] == 0     # offset at 8 s
    assert sf[200, 0] == 1                       # second event


def test_stimfunction_weights_and_overlap():
    sf = sim.generate_stimfunction(onsets=[0, 2], event_durations=[4],
                                   total_time=10, weights=[1, 3],
                                   temporal_resolution=1)
    # overlapping events: the later weight wins or accumulates —
    # reference keeps the later assignment
    assert sf[0, 0] == 1
    assert sf[3, 0] == 3


def test_export_3_column_round_trip(tmp_path):
    sf = sim.generate_stimfunction(onsets=[2, 9], event_durations=[3],
                                   total_time=20,
                                   temporal_resolution=10)
    f = tmp_path / "ev3.txt"
    sim.export_3_column(sf, str(f), temporal_resolution=10)
    rows = np.loadtxt(str(f)).reshape(-1, 3)
    assert rows.shape[0] == 2
    assert np.allclose(rows[:, 0], [2.0, 9.0], atol=0.11)
    assert np.allclose(rows[:, 1], 3.0, atol=0.11)


def test_export_epoch_file_round_trip(tmp_path):
    sf = sim.generate_stimfunction(onsets=[0, 10], event_durations=[5],
                                   total_time=20,
                                   temporal_resolution=1)
    f = tmp_path / "epochs.npy"
    sim.export_epoch_file([sf], str(f), tr_duration=2,
                          temporal_resolution=1)
    ep = np.load(str(f), allow_pickle=True)
    # per-subject object array of [conditions, epochs, TRs] one-hots
    arr = np.asarray(ep[0])
    assert arr.shape == (1, 2, 10)        # 1 cond, 2 epochs, 20s/2s TR
    assert arr.dtype == bool
    # first epoch spans TRs 0-2, second starts at TR 5
    assert arr[0, 0, 0] and arr[0, 0, 2] and not arr[0, 0, 5]
    assert arr[0, 1, 5] and arr[0, 1, 7]


# -- HRF ----------------------------------------------------------------------

def test_double_gamma_hrf_properties():
    hrf = sim._double_gamma_hrf(temporal_resolution=10)
    hrf = np.asarray(hrf)
    peak = np.argmax(hrf) / 10.0
    assert 4 < peak < 8                    # canonical ~6 s peak
    trough = np.argmin(hrf) / 10.0
    assert 10 < trough < 20                # undershoot after the peak
    assert hrf.max() > 0 > hrf.min()
    assert abs(hrf[-1]) < 0.1 * hrf.max()  # decays back toward zero


def test_convolve_hrf_shapes_and_scaling():
    sf = sim.generate_stimfunction(onsets=[4], event_durations=[2],
                                   total_time=40,
                                   temporal_resolution=10)
    conv = sim.convolve_hrf(sf, tr_duration=2, temporal_resolution=10,
                            scale_function=True)
    assert conv.shape == (20, 1)
    assert np.isclose(conv.max(), 1.0)
    # response peaks AFTER the stimulus onset (2 TRs = 4 s + HRF lag)
    assert conv[:2].max() < 0.5


def test_apply_signal_outer_product():
    sf = np.zeros((10, 1))
    sf[3:6] = 1.0
    vol = np.zeros((4, 4, 4))
    vol[1, 2, 3] = 2.0
    out = sim.apply_signal(sf, vol)
    assert out.shape == (4, 4, 4, 10)
    assert out[1, 2, 3, 4] == 2.0
    assert out[1, 2, 3, 0] == 0.0
    assert out[0, 0, 0].max() == 0.0


# -- noise components ---------------------------------------------------------

def test_generate_noise_spatial_smoothness():
    np.random.seed(3)
    rough = sim._generate_noise_spatial(np.array([24, 24, 24]), fwhm=1.0)
    smooth = sim._generate_noise_spatial(np.array([24, 24, 24]),
                                         fwhm=8.0)
    # lag-1 spatial autocorrelation rises with fwhm
    def lag1(v):
        return np.corrcoef(v[:-1].ravel(), v[1:].ravel())[0, 1]
    assert lag1(smooth) > lag1(rough)


def test_generate_noise_system_dims():
    np.random.seed(4)
    n = sim._generate_noise_system(np.array([8, 8, 8, 20]), 1.0, 1.0)
    assert n.shape == (8, 8, 8, 20)
    assert np.isfinite(n).all()


def test_temporal_drift_is_slow():
    np.random.seed(5)
    drift = sim._generate_noise_temporal_drift(100, 2.0)
    drift = np.asarray(drift).ravel()
    assert drift.shape[0] == 100
    # drift has most of its power at low frequency
    p = np.abs(np.fft.rfft(drift - drift.mean())) ** 2
    assert p[1:5].sum() > p[5:].sum()


def test_autoregression_rho_controls_smoothness():
    np.random.seed(6)
    dims = np.array([4, 4, 4])
    mask = np.ones(dims)
    t = list(range(200))
    nd_hi = {'auto_reg_rho': [0.9], 'ma_rho': [0.0], 'fwhm': 4}
    nd_lo = {'auto_reg_rho': [0.1], 'ma_rho': [0.0], 'fwhm': 4}
    hi = sim._generate_noise_temporal_autoregression(t, nd_hi, dims,
                                                     mask)
    lo = sim._generate_noise_temporal_autoregression(t, nd_lo, dims,
                                                     mask)

    def ac1(v):
        x = v[2, 2, 2] - v[2, 2, 2].mean()
        return np.dot(x[:-1], x[1:]) / np.dot(x, x)
    assert ac1(hi) > ac1(lo) + 0.3
    # MA order > AR order must raise
    with pytest.raises(ValueError):
        sim._generate_noise_temporal_autoregression(
            t, {'auto_reg_rho': [0.5], 'ma_rho': [0.1, 0.2], 'fwhm': 4},
            dims, mask)


def test_calc_fwhm_tracks_smoothing():
    np.random.seed(7)
    dims = np.array([20, 20, 20])
    mask = np.ones(dims)
    rough = sim._generate_noise_spatial(dims, fwhm=1.0)
    smooth = sim._generate_noise_spatial(dims, fwhm=6.0)
    f_r = sim._calc_fwhm(rough, mask, [1.0, 1.0, 1.0])
    f_s = sim._calc_fwhm(smooth, mask, [1.0, 1.0, 1.0])
    assert f_s > f_r


def test_generate_noise_matched_components(tmp_path):
    """generate_noise honors sfnr/snr targets loosely even unmatched,
    and produces brain >> nonbrain intensity."""
    np.random.seed(8)
    dims = np.array([12, 12, 12])
    stim = np.zeros(30)
    template = np.zeros(dims)
    template[3:9, 3:9, 3:9] = 0.9
    mask = (template > 0).astype(float)
    nd = {'sfnr': 80, 'snr': 40, 'max_activity': 1000, 'matched': 0}
    noise = sim.generate_noise(dims, stim, 2.0, template, mask,
                               noise_dict=dict(nd))
    brain = noise[mask > 0].mean()
    nonbrain = noise[mask == 0].mean()
    assert brain > 5 * max(nonbrain, 1e-9)


# -- signal scaling -----------------------------------------------------------

@pytest.mark.parametrize("method,expected", [
    ("SFNR", 200 / 50.0),
    ("PSC", 2.0),
])
def test_compute_signal_change_methods(method, expected):
    sf = np.vstack([np.sin(np.linspace(0, 6, 50))]).T
    noise = np.ones((50, 1)) * 200
    nd = {'sfnr': 50}
    scaled = sim.compute_signal_change(sf.copy(), noise, nd, [1.0],
                                       method=method)
    assert np.isclose(np.max(np.abs(scaled)), expected, rtol=1e-6)


def test_compute_signal_change_multi_magnitude():
    sf = np.hstack([np.ones((30, 1)), np.ones((30, 1))])
    noise = np.ones((30, 2)) * 100
    nd = {'sfnr': 50}
    out = sim.compute_signal_change(sf.copy(), noise, nd, [1.0, 3.0],
                                    method='PSC')
    assert np.isclose(out[:, 1].max() / out[:, 0].max(), 3.0)


# -- 1-D receptive fields -----------------------------------------------------

def test_1d_rf_tuning_curves():
    rfs, tuning = sim.generate_1d_gaussian_rfs(
        n_voxels=30, feature_resolution=90, feature_range=(0, 90),
        rf_size=10.0, random_tuning=False)
    assert rfs.shape == (30, 90)
    assert np.allclose(rfs.max(axis=1), 1.0)
    trial_list = np.array([10, 45, 80])
    resp = sim.generate_1d_rf_responses(rfs, trial_list,
                                        feature_resolution=90,
                                        feature_range=(0, 90),
                                        trial_noise=0.0)
    assert resp.shape == (30, 3)
    assert np.isfinite(resp).all()
    # the voxel tuned to the trial's feature responds maximally
    for t, feat in enumerate(trial_list):
        best_vox = np.argmax(resp[:, t])
        assert abs(tuning[best_vox] - feat) <= 6


def test_compute_signal_change_all_methods():
    """Every method's scaling rule against its closed form
    (ref fmrisim.py:3072 semantics)."""
    rng = np.random.RandomState(0)
    T, V = 60, 2
    sf = np.abs(rng.randn(T, V)) + 0.1
    noise = rng.randn(T, V) * 3 + 100.0
    nd = {'sfnr': 50.0}
    sfn = sf / np.max(np.abs(sf))
    for method, expect in [
        ('SFNR', lambda s, n, m: s * (n.mean() / nd['sfnr'] * m)),
        ('CNR_Amp/Noise-SD', lambda s, n, m: s * (m * n.std())),
        ('PSC', lambda s, n, m: s * (n.mean() / 100 * m)),
        ('CNR_Signal-SD/Noise-SD',
         lambda s, n, m: s * ((m / np.max(np.abs(s))) * n.std()
                              / s.std())),
    ]:
        out = sim.compute_signal_change(sf.copy(), noise, nd,
                                        [1.5, 1.5], method=method)
        for v in range(V):
            want = expect(sfn[:, v], noise[:, v], 1.5)
            assert np.allclose(out[:, v], want), method
    # dB variants scale with 10^(mag/20)
    out_db = sim.compute_signal_change(sf.copy(), noise, nd, [20.0],
                                       method='CNR_Amp2/Noise-Var_dB')
    out_db0 = sim.compute_signal_change(sf.copy(), noise, nd, [0.0],
                                        method='CNR_Amp2/Noise-Var_dB')
    assert np.allclose(out_db, out_db0 * 10.0)
    with pytest.raises(ValueError):
        sim.compute_signal_change(sf, noise, nd, [1.0],
                                  method='nonsense')
