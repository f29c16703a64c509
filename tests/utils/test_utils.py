import math

import numpy as np
import pytest

from brainiak_amd.utils import utils


def test_circ_dist():
    x = np.array([0.1, 2 * np.pi - 0.1, np.pi])
    y = np.array([2 * np.pi - 0.1, 0.1, -np.pi])
    r = utils.circ_dist(x, y)
    assert np.allclose(r, [0.2, -0.2, 0.0], atol=1e-12)
    with pytest.raises(ValueError):
        utils.circ_dist(np.zeros(3), np.zeros(4))


def test_tri_sym_roundtrip(seeded_rng):
    dim = 5
    sym = seeded_rng.rand(dim, dim)
    sym = sym + sym.T
    tri = utils.from_sym_2_tri(sym)
    assert tri.shape == (dim * (dim + 1) // 2,)
    back = utils.from_tri_2_sym(tri, dim)
    assert np.allclose(np.triu(back), np.triu(sym))


def test_sumexp_stable(seeded_rng):
    data = seeded_rng.randn(4, 3) * 50
    s, m, e = utils.sumexp_stable(data)
    expected = np.exp(data - data.max(axis=0)).sum(axis=0)
    assert np.allclose(s, expected)
    assert np.allclose(m, data.max(axis=0))
    assert e.shape == data.shape
    assert np.all(np.isfinite(e))


def test_concatenate_not_none():
    arrays = [None, np.ones((2, 2)), None, np.zeros((3, 2))]
    out = utils.concatenate_not_none(arrays, axis=0)
    assert out.shape == (5, 2)
    assert np.allclose(out[:2], 1) and np.allclose(out[2:], 0)


def test_cov2corr():
    cov = np.array([[4.0, 2.0], [2.0, 9.0]])
    corr = utils.cov2corr(cov)
    assert np.allclose(np.diag(corr), 1.0)
    assert np.allclose(corr[0, 1], 2.0 / 6.0)


def test_center_mass_exp():
    # full support → mean = scale
    assert math.isclose(utils.center_mass_exp((0, np.inf), scale=2.0), 2.0)
    # narrow interval → approximately its midpoint
    m = utils.center_mass_exp((1.0, 1.001), scale=1.0)
    assert 1.0 < m < 1.001
    with pytest.raises(AssertionError):
        utils.center_mass_exp((-1.0, 1.0))


def test_usable_cpu_count():
    assert utils.usable_cpu_count() >= 1


def test_phase_randomize_preserves_spectrum(seeded_rng):
    data = seeded_rng.randn(60, 4, 3)
    shifted = utils.phase_randomize(data, random_state=0)
    assert shifted.shape == data.shape
    # power spectrum magnitude preserved per voxel/subject
    f0 = np.abs(np.fft.fft(data, axis=0))
    f1 = np.abs(np.fft.fft(shifted, axis=0))
    assert np.allclose(f0, f1, atol=1e-8)
    # but the time series differ
    assert not np.allclose(data, shifted)
    # 2-D input keeps 2-D output
    d2 = seeded_rng.randn(40, 3)
    s2 = utils.phase_randomize(d2, random_state=1)
    assert s2.shape == d2.shape


def test_phase_randomize_voxelwise(seeded_rng):
    data = seeded_rng.randn(31, 5, 2)
    s_same = utils.phase_randomize(data, voxelwise=False, random_state=0)
    s_vox = utils.phase_randomize(data, voxelwise=True, random_state=0)
    assert s_same.shape == s_vox.shape == data.shape
    corr_same = np.corrcoef(s_same[:, 0, 0], s_same[:, 1, 0])[0, 1]
    assert np.isfinite(corr_same)


def test_p_from_null():
    dist = np.arange(-49.5, 50.5)  # 100 samples
    p_right = utils.p_from_null(40.0, dist, side='right')
    # 10 samples >= 40 → (10+1)/(100+1)
    assert math.isclose(p_right, 11 / 101)
    p_exact = utils.p_from_null(40.0, dist, side='right', exact=True)
    assert math.isclose(p_exact, 0.1)
    p_two = utils.p_from_null(49.5, dist, side='two-sided')
    assert math.isclose(p_two, 3 / 101)  # |−49.5| and |49.5| both match
    with pytest.raises(ValueError):
        utils.p_from_null(0, dist, side='bogus')


def test_array_correlation(seeded_rng):
    x = seeded_rng.randn(50, 4)
    y = seeded_rng.randn(50, 4)
    r = utils.array_correlation(x, y)
    expected = [np.corrcoef(x[:, i], y[:, i])[0, 1] for i in range(4)]
    assert np.allclose(r, expected)
    # axis=1: rows as variables
    r_rows = utils.array_correlation(x.T, y.T, axis=1)
    assert np.allclose(r_rows, expected)
    # perfect correlation
    assert np.allclose(utils.array_correlation(x[:, 0], 2 * x[:, 0] + 1), 1.0)
    with pytest.raises(ValueError):
        utils.array_correlation(np.zeros((3, 2)), np.zeros((4, 2)))


def test_gen_design_fsl(tmp_path):
    stim = tmp_path / "cond.txt"
    np.savetxt(stim, np.array([[10.0, 2.0, 1.0], [40.0, 2.0, 1.0]]))
    design = utils.gen_design([str(stim)], scan_duration=80.0, TR=2.0,
                              style='FSL')
    assert design.shape == (40, 1)
    # the HRF response should peak a few TRs after each onset
    assert design[:5].max() < 1e-6
    peak1 = np.argmax(design[:15, 0])
    assert 6 <= peak1 * 2 <= 22
    assert design.max() > 0


def test_gen_design_multirun(tmp_path):
    stim = tmp_path / "cond.txt"
    # onset 70 lands in the second run (runs of 60s each)
    np.savetxt(stim, np.array([[5.0, 1.0, 1.0], [70.0, 1.0, 1.0]]))
    design = utils.gen_design([str(stim)], scan_duration=[60.0, 60.0],
                              TR=2.0, style='FSL')
    assert design.shape == (60, 1)
    # response in both runs
    assert design[:30].max() > 0
    assert design[30:].max() > 0
    # no leakage: start of run 2 is clean until after onset 10s into it
    assert design[30:33].max() < design[30:].max() * 0.2


def test_read_design_empty():
    rd = utils.ReadDesign()
    assert rd.n_col == 0
    assert rd.design_task.shape[1] == 0 or rd.design_task.size == 0
