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


# -- round-2 depth (ref tests/utils/test_utils.py:1-455) ---------------------

def test_read_design_afni_file(tmp_path):
    """ReadDesign parses an AFNI-style .1D design matrix with header
    comments (the reference's AFNI reader contract)."""
    from brainiak_amd.utils.utils import ReadDesign
    content = (
        '# <matrix\n'
        '#  ni_type = "4*double"\n'
        '#  ColumnLabels = "Run#1Pol#0 ; Run#1Pol#1 ; taskA#0 ; '
        'taskB#0"\n'
        '# >\n'
        '1 0.1 0 0\n'
        '1 0.2 1 0\n'
        '1 0.3 0 1\n'
        '1 0.4 0 0\n')
    p = tmp_path / "design.1D"
    p.write_text(content)
    d = ReadDesign(fname=str(p), include_orth=False, include_pols=False)
    assert d.design_task.shape[0] == 4
    assert d.n_TR == 4


def test_gen_design_afni_style(tmp_path):
    from brainiak_amd.utils.utils import gen_design
    f = tmp_path / "times.txt"
    # AFNI style: one row per scan, '*' for empty scans
    f.write_text("2.0 10.0\n*\n")
    design = gen_design([str(f)], scan_duration=[30.0, 30.0], TR=2.0,
                        style='AFNI')
    assert design.shape == (30, 1)
    assert design.max() > 0
    # second scan has no events: its rows stay near zero after the
    # HRF from scan 1 cannot leak across the scan boundary
    assert np.allclose(design[15 + 10:], 0, atol=1e-3)


def test_gen_design_fsl_durations_and_weights(tmp_path):
    from brainiak_amd.utils.utils import gen_design
    f = tmp_path / "ev.txt"
    f.write_text("4.0 2.0 1.0\n12.0 2.0 2.0\n")
    d1 = gen_design([str(f)], scan_duration=40.0, TR=1.0)
    assert d1.shape == (40, 1)
    # the weight-2 event drives ~2x the response of the weight-1 event
    peak1 = d1[4:12].max()
    peak2 = d1[12:24].max()
    assert 1.5 < peak2 / peak1 < 2.5


def test_phase_randomize_reproducible(seeded_rng):
    from brainiak_amd.utils.utils import phase_randomize
    data = seeded_rng.randn(32, 4, 3)
    a = phase_randomize(data, random_state=7)
    b = phase_randomize(data, random_state=7)
    c = phase_randomize(data, random_state=8)
    assert np.array_equal(a, b)
    assert not np.array_equal(a, c)


def test_p_from_null_exact_mode(seeded_rng):
    """exact=True drops the +1 correction (the reference's exact-test
    branch, utils.py:862-874)."""
    from brainiak_amd.utils.utils import p_from_null
    null = np.arange(-10.0, 10.0)
    p_ex = p_from_null(np.array(100.0), null, side='right', exact=True)
    p_mc = p_from_null(np.array(100.0), null, side='right', exact=False)
    assert p_ex == 0.0                 # exact: can be zero
    assert p_mc > 0.0                  # Monte-Carlo: floored


def test_sumexp_stable_extremes():
    from brainiak_amd.utils.utils import sumexp_stable
    data = np.array([[1000.0, -1000.0], [1001.0, -999.0]])
    s, m, ex = sumexp_stable(data)
    assert np.all(np.isfinite(s))
    assert np.allclose(m, [1001.0, -999.0])


def test_center_mass_exp_errors():
    from brainiak_amd.utils.utils import center_mass_exp
    with pytest.raises(AssertionError):
        center_mass_exp((1.0, 0.5))     # right <= left
    with pytest.raises(AssertionError):
        center_mass_exp((-1.0, 1.0))    # negative support
    with pytest.raises(AssertionError):
        center_mass_exp([0.0, 1.0])     # not a tuple
    # half-open interval reduces to left + scale
    assert center_mass_exp((2.0, np.inf), scale=3.0) == 5.0


def test_array_correlation_matches_corrcoef(seeded_rng):
    from brainiak_amd.utils.utils import array_correlation
    x = seeded_rng.randn(50, 7)
    y = seeded_rng.randn(50, 7)
    r = array_correlation(x, y, axis=0)
    for j in range(7):
        assert np.isclose(r[j], np.corrcoef(x[:, j], y[:, j])[0, 1])
    # axis=1 path
    r1 = array_correlation(x.T, y.T, axis=1)
    assert np.allclose(r1, r)
    with pytest.raises(ValueError):
        array_correlation(x, y[:10])


def test_p_from_null_sides_and_validation(seeded_rng):
    """left/right/two-sided relationships and the invalid-side error
    (ref tests/utils/test_utils.py:184-197)."""
    null = np.arange(100, dtype=float)
    p_r = utils.p_from_null(80.0, null, side='right')
    p_l = utils.p_from_null(80.0, null, side='left')
    p_t = utils.p_from_null(80.0, null, side='two-sided')
    # Phipson-Smyth smoothing: (count + 1) / (n + 1)
    assert np.isclose(p_r, (19 + 1 + 1) / 101)   # >= 80: 80..99 + obs
    assert p_l > p_r
    assert p_t <= 2 * min(p_l, p_r) + 1e-12
    with pytest.raises(ValueError):
        utils.p_from_null(1.0, null, side='wrong')
    # exact mode skips the +1 smoothing
    p_exact = utils.p_from_null(80.0, null, side='right', exact=True)
    assert p_exact < p_r
