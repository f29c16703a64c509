"""General-purpose helpers (API parity: reference src/brainiak/utils/utils.py).

Pure numpy/scipy; nothing here is a GPU hot path.  Hot-path math lives in
``brainiak_amd.ops``.  Functions:

- ``circ_dist``            (ref utils.py:48)
- ``from_tri_2_sym`` / ``from_sym_2_tri``   (ref utils.py:69/95)
- ``sumexp_stable``        (ref utils.py:118)
- ``concatenate_not_none`` (ref utils.py:154)
- ``cov2corr``             (ref utils.py:185)
- ``ReadDesign``           (ref utils.py:208) — AFNI 3dDeconvolve .1D reader
- ``gen_design``           (ref utils.py:365) — FSL/AFNI stimulus timing → design
- ``center_mass_exp``      (ref utils.py:657)
- ``usable_cpu_count``     (ref utils.py:701)
- ``phase_randomize``      (ref utils.py:720)
- ``p_from_null``          (ref utils.py:803) — Phipson & Smyth +1 correction
- ``array_correlation``    (ref utils.py:938)
"""

import logging
import os
import re
import warnings

import numpy as np
from scipy.fftpack import fft, ifft

logger = logging.getLogger(__name__)

__all__ = [
    "ReadDesign",
    "array_correlation",
    "center_mass_exp",
    "circ_dist",
    "concatenate_not_none",
    "cov2corr",
    "from_sym_2_tri",
    "from_tri_2_sym",
    "gen_design",
    "p_from_null",
    "phase_randomize",
    "sumexp_stable",
    "usable_cpu_count",
]


def circ_dist(x, y):
    """Pairwise circular distance (radians) between two equal-size arrays."""
    x = np.asarray(x)
    y = np.asarray(y)
    if x.size != y.size:
        raise ValueError(
            "Input sizes must match to compute pairwise comparisons.")
    # angle of the ratio of unit phasors = signed circular difference
    return np.angle(np.exp(1j * x) * np.exp(-1j * y))


def from_tri_2_sym(tri, dim):
    """Unpack a 1-D upper-triangle vector into a [dim, dim] symmetric matrix.

    Only the upper triangle is filled (matching the reference: the lower
    triangle stays zero).
    """
    symm = np.zeros((dim, dim))
    symm[np.triu_indices(dim)] = tri
    return symm


def from_sym_2_tri(symm):
    """Pack the upper triangle (incl. diagonal) of a matrix into 1-D."""
    symm = np.asarray(symm)
    return symm[np.triu_indices_from(symm)]


def sumexp_stable(data):
    """Numerically-stable sum of exponentials per column.

    Returns (result_sum, max_value, result_exp) with ``data`` shaped
    [features, samples]: exp is taken after subtracting each column max.
    """
    data = np.asarray(data)
    max_value = data.max(axis=0)
    result_exp = np.exp(data - max_value)
    result_sum = np.sum(result_exp, axis=0)
    return result_sum, max_value, result_exp


def concatenate_not_none(data, axis=0):
    """Concatenate the non-None entries of a list of arrays."""
    return np.concatenate([d for d in data if d is not None], axis=axis)


def cov2corr(cov):
    """Convert a covariance matrix to a correlation matrix."""
    cov = np.asarray(cov)
    assert cov.ndim == 2, 'covariance matrix should be 2D array'
    inv_sd = 1.0 / np.sqrt(np.diag(cov))
    return cov * inv_sd[None, :] * inv_sd[:, None]


class ReadDesign:
    """Reader for AFNI 3dDeconvolve design matrices (.1D/.1d/.txt).

    Same attribute contract as the reference class (ref utils.py:208):
    ``design``, ``design_task``, ``n_col``, ``column_types`` (1 = task,
    0 = orthogonal/motion, -1 = polynomial drift), ``n_basis``, ``n_stim``,
    ``n_orth``, ``StimLabels``, ``reg_nuisance``, ``n_TR``.
    """

    def __init__(self, fname=None, include_orth=True, include_pols=True):
        if fname is None:
            self.design = np.zeros([0, 0])
            self.n_col = 0
            self.column_types = np.ones(0)
            self.n_basis = 0
            self.n_stim = 0
            self.n_orth = 0
            self.StimLabels = []
        else:
            _, ext = os.path.splitext(fname)
            if ext in ('.1D', '.1d', '.txt'):
                self.read_afni(fname)

        self.include_orth = include_orth
        self.include_pols = include_pols
        self.cols_task = np.where(self.column_types == 1)[0]
        self.design_task = self.design[:, self.cols_task]
        if self.design_task.ndim == 1:
            self.design_task = self.design_task[:, None]
        self.n_TR = self.design_task.shape[0]

        nuis_cols = []
        if self.include_orth:
            nuis_cols.append(np.where(self.column_types == 0)[0])
        if self.include_pols:
            nuis_cols.append(np.where(self.column_types == -1)[0])
        cols = np.sort(np.concatenate(nuis_cols)) if nuis_cols else \
            np.array([], dtype=np.intp)
        self.cols_nuisance = np.intp(cols)
        if self.cols_nuisance.size > 0:
            self.reg_nuisance = self.design[:, self.cols_nuisance]
            if self.reg_nuisance.ndim == 1:
                self.reg_nuisance = self.reg_nuisance[:, None]
        else:
            self.reg_nuisance = None

    def read_afni(self, fname):
        self.n_basis = 0
        self.n_stim = 0
        self.n_orth = 0
        self.StimLabels = []
        self.design = np.loadtxt(fname, ndmin=2)
        with open(fname) as f:
            text = f.read()

        m = re.search(r'^#[ ]+ni_type[ ]+=[ ]+"(\d+)[*]', text, re.MULTILINE)
        if m:
            self.n_col = int(m.group(1))
            if self.n_col != self.design.shape[1]:
                warnings.warn('The number of columns in the design matrix'
                              'does not match the header information')
                self.n_col = self.design.shape[1]
        else:
            self.n_col = self.design.shape[1]

        self.column_types = np.ones(self.n_col)
        m = re.search(r'^#[ ]+ColumnGroups[ ]+=[ ]+"(.+)"', text, re.MULTILINE)
        if m:
            idx = 0
            for group in m.group(1).split(','):
                at_parts = group.split('@')
                if len(at_parts) == 2:
                    # "<count>@<type>"
                    n = int(at_parts[0])
                    self.column_types[idx:idx + n] = int(at_parts[1])
                    idx += n
                elif len(at_parts) == 1 and not re.search(r'\..', at_parts[0]):
                    self.column_types[idx] = int(at_parts[0])
                    idx += 1
                else:
                    # "a..b" range form: a run of task-condition columns
                    n = int(re.split(r'\..', group)[1])
                    self.column_types[idx:idx + n] = 1
                    idx += n
            self.n_basis = np.sum(self.column_types == -1)
            self.n_stim = np.sum(self.column_types > 0)
            self.n_orth = np.sum(self.column_types == 0)

        m = re.search(r'^#[ ]+StimLabels[ ]+=[ ]+"(.+)"', text, re.MULTILINE)
        self.StimLabels = re.split(r'[ ;]+', m.group(1)) if m else []


def _double_gamma_hrf_curve(t, response_delay=6, undershoot_delay=12,
                            response_dispersion=0.9,
                            undershoot_dispersion=0.9,
                            undershoot_scale=0.035):
    """Double-gamma HRF sampled at times ``t`` (seconds), unnormalized."""
    from scipy.stats import gamma
    resp = gamma.pdf(t, response_delay / response_dispersion,
                     loc=0, scale=response_dispersion)
    under = gamma.pdf(t, undershoot_delay / undershoot_dispersion,
                      loc=0, scale=undershoot_dispersion)
    return resp - undershoot_scale * under


def _parse_stimtime_fsl(stimtime_files, n_C, n_S, scan_onoff):
    """FSL 3-column (onset, duration, weight) event files → per-scan info."""
    info = [[{'onset': [], 'duration': [], 'weight': []}
             for _ in range(n_C)] for _ in range(n_S)]
    for i_c, fname in enumerate(stimtime_files):
        content = np.loadtxt(fname, ndmin=2)
        for row in content:
            onset = float(row[0])
            duration = float(row[1]) if row.size > 1 else 1.0
            weight = float(row[2]) if row.size > 2 else 1.0
            # locate the scan containing this onset
            i_s = int(np.searchsorted(scan_onoff[1:], onset, side='right'))
            if i_s >= n_S:
                continue
            info[i_s][i_c]['onset'].append(onset - scan_onoff[i_s])
            info[i_s][i_c]['duration'].append(duration)
            info[i_s][i_c]['weight'].append(weight)
    return info


def _parse_stimtime_afni(stimtime_files, n_C, n_S, scan_onoff):
    """AFNI one-line-per-scan ``onset*weight:duration`` files."""
    info = [[{'onset': [], 'duration': [], 'weight': []}
             for _ in range(n_C)] for _ in range(n_S)]
    for i_c, fname in enumerate(stimtime_files):
        with open(fname) as f:
            lines = f.read().splitlines()
        for i_s, line in enumerate(lines):
            if i_s >= n_S:
                break
            scan_len = scan_onoff[i_s + 1] - scan_onoff[i_s]
            for tok in line.split():
                if tok == '*':
                    continue
                m = re.match(
                    r'^(?P<onset>[-0-9.eE+]+)'
                    r'([*](?P<weight>[-0-9.eE+]+))?'
                    r'([:](?P<duration>[-0-9.eE+]+))?$', tok)
                if not m:
                    continue
                onset = float(m.group('onset'))
                if onset < 0 or onset >= scan_len:
                    continue
                info[i_s][i_c]['onset'].append(onset)
                info[i_s][i_c]['weight'].append(
                    float(m.group('weight')) if m.group('weight') else 1.0)
                info[i_s][i_c]['duration'].append(
                    float(m.group('duration')) if m.group('duration') else 1.0)
    return info


def gen_design(stimtime_files, scan_duration, TR, style='FSL',
               temp_res=0.01,
               hrf_para={'response_delay': 6, 'undershoot_delay': 12,
                         'response_dispersion': 0.9,
                         'undershoot_dispersion': 0.9,
                         'undershoot_scale': 0.035}):
    """Generate a [n_TRs, n_conditions] design matrix from stimulus timing
    files (FSL 3-column or AFNI style), convolved with a double-gamma HRF.

    Contract matches ref utils.py:365 (multi-run concatenation; onsets in
    FSL files are relative to the start of the first run but responses do
    not leak across run boundaries; the design is scaled by ``temp_res``).
    """
    if np.ndim(scan_duration) == 0:
        scan_duration = [scan_duration]
    scan_duration = np.asarray(scan_duration, dtype=np.float64)
    assert np.all(scan_duration > TR), \
        'scan duration should be longer than a TR'
    if isinstance(stimtime_files, str):
        stimtime_files = [stimtime_files]
    assert TR > 0, 'TR should be positive'
    assert style in ('FSL', 'AFNI'), 'style can only be FSL or AFNI'

    n_C = len(stimtime_files)
    n_S = scan_duration.size
    scan_onoff = np.insert(np.cumsum(scan_duration), 0, 0)
    if style == 'FSL':
        info = _parse_stimtime_fsl(stimtime_files, n_C, n_S, scan_onoff)
    else:
        info = _parse_stimtime_afni(stimtime_files, n_C, n_S, scan_onoff)

    designs = []
    for i_s in range(n_S):
        n_TR = int(np.round(scan_duration[i_s] / TR))
        n_fine = int(np.round(scan_duration[i_s] / temp_res))
        design = np.zeros((n_TR, n_C))
        # HRF on the fine grid (support out to ~32 s is plenty)
        hrf_t = np.arange(0, 32.0, temp_res)
        hrf = _double_gamma_hrf_curve(hrf_t, **hrf_para)
        for i_c in range(n_C):
            ev = info[i_s][i_c]
            if len(ev['onset']) == 0:
                continue
            boxcar = np.zeros(n_fine)
            for onset, dur, w in zip(ev['onset'], ev['duration'],
                                     ev['weight']):
                a = int(np.round(onset / temp_res))
                b = int(np.round((onset + dur) / temp_res))
                boxcar[a:min(b, n_fine)] = w
            resp = np.convolve(boxcar, hrf)[:n_fine]
            # sample at TR boundaries; scale by temp_res so the design is
            # independent of the fine-grid resolution
            tr_idx = (np.arange(n_TR) * (TR / temp_res)).astype(np.intp)
            design[:, i_c] = resp[np.minimum(tr_idx, n_fine - 1)] * temp_res
        designs.append(design)
    return np.concatenate(designs, axis=0)


def center_mass_exp(interval, scale=1.0):
    """Center of mass of exp(-x/scale)/scale restricted to ``interval``."""
    assert isinstance(interval, tuple), 'interval must be a tuple'
    assert len(interval) == 2, 'interval must be length two'
    left, right = interval
    assert left >= 0, 'interval_left must be non-negative'
    assert right > left, \
        'interval_right must be bigger than interval_left'
    assert scale > 0, 'scale must be positive'
    if right < np.inf:
        num = ((left + scale) * np.exp(-left / scale)
               - (scale + right) * np.exp(-right / scale))
        den = np.exp(-left / scale) - np.exp(-right / scale)
        return num / den
    return left + scale


def usable_cpu_count():
    """CPUs usable by the current process (honours cpuset affinity)."""
    try:
        return len(os.sched_getaffinity(0))
    except AttributeError:  # pragma: no cover - non-Linux fallback
        return os.cpu_count()


def _check_timeseries_input(data):
    """Standardize time-series input to [n_TRs, n_voxels, n_subjects]."""
    if isinstance(data, list):
        shape0 = data[0].shape
        for i, d in enumerate(data):
            if d.shape != shape0:
                raise ValueError(
                    "All ndarrays in input list must be the same shape!")
            if d.ndim == 1:
                data[i] = d[:, np.newaxis]
        data = np.dstack(data)
    elif isinstance(data, np.ndarray):
        if data.ndim == 2:
            data = data[:, np.newaxis, :]
        elif data.ndim != 3:
            raise ValueError("Input ndarray should have 2 "
                             "or 3 dimensions (got {0})!".format(data.ndim))
    n_TRs, n_voxels, n_subjects = data.shape
    logger.info("Assuming %d subjects with %d time points and %d voxel(s) "
                "or ROI(s) for ISC analysis.", n_subjects, n_TRs, n_voxels)
    return data, n_TRs, n_voxels, n_subjects


def phase_randomize(data, voxelwise=False, random_state=None):
    """Phase-randomize time series per subject (FFT phase shuffling).

    Positive and negative frequencies are shifted symmetrically so the
    output stays real; power spectrum (hence autocorrelation) is preserved.
    """
    data_ndim = np.ndim(data)
    data, n_TRs, n_voxels, n_subjects = _check_timeseries_input(data)
    prng = (random_state if isinstance(random_state, np.random.RandomState)
            else np.random.RandomState(random_state))

    if n_TRs % 2 == 0:
        pos_freq = np.arange(1, n_TRs // 2)
        neg_freq = np.arange(n_TRs - 1, n_TRs // 2, -1)
    else:
        pos_freq = np.arange(1, (n_TRs - 1) // 2 + 1)
        neg_freq = np.arange(n_TRs - 1, (n_TRs - 1) // 2, -1)

    if not voxelwise:
        phase_shifts = prng.rand(len(pos_freq), 1, n_subjects) * 2 * np.pi
    else:
        phase_shifts = (prng.rand(len(pos_freq), n_voxels, n_subjects)
                        * 2 * np.pi)

    fft_data = fft(data, axis=0)
    fft_data[pos_freq, :, :] *= np.exp(1j * phase_shifts)
    fft_data[neg_freq, :, :] *= np.exp(-1j * phase_shifts)
    shifted = np.real(ifft(fft_data, axis=0))
    if data_ndim == 2:
        shifted = shifted[:, 0, :]
    return shifted


def p_from_null(observed, distribution, side='two-sided', exact=False,
                axis=None):
    """p-value of ``observed`` against a resampled null distribution.

    Non-exact tests use the Phipson & Smyth (2010) +1 correction so a
    randomly-drawn permutation p-value is never zero.
    """
    if side not in ('two-sided', 'left', 'right'):
        raise ValueError("The value for 'side' must be either "
                         "'two-sided', 'left', or 'right', got {0}".
                         format(side))
    n_samples = len(distribution)
    logger.info("Assuming %d resampling iterations", n_samples)
    if side == 'two-sided':
        numerator = np.sum(np.abs(distribution) >= np.abs(observed),
                           axis=axis)
    elif side == 'left':
        numerator = np.sum(distribution <= observed, axis=axis)
    else:
        numerator = np.sum(distribution >= observed, axis=axis)
    if exact:
        return numerator / n_samples
    return (numerator + 1) / (n_samples + 1)


def array_correlation(x, y, axis=0):
    """Column- (axis=0) or row-wise (axis=1) Pearson correlation of two
    equal-shape arrays."""
    x = np.asarray(x)
    y = np.asarray(y)
    if x.shape != y.shape:
        raise ValueError("Input arrays must be the same shape")
    if axis == 1:
        x, y = x.T, y.T
    xd = x - np.mean(x, axis=0)
    yd = y - np.mean(y, axis=0)
    numerator = np.sum(xd * yd, axis=0)
    denominator = np.sqrt(np.sum(xd ** 2, axis=0) * np.sum(yd ** 2, axis=0))
    return numerator / denominator
