"""fMRI simulator (API parity: ref src/brainiak/utils/fmrisim.py).

Generates realistic synthetic fMRI volumes: signal features of
parametric shapes, stimulus boxcars convolved with a double-gamma HRF,
and noise with matched SFNR/SNR/FWHM/AR(MA) structure (system, drift,
physiological, task and autoregressive components), plus noise
*estimation* (``calc_noise``) so simulated participants can be matched
to real ones.

Deviations from the reference:
 - ARMA coefficients are estimated by a batched conditional-MLE
   (torch L-BFGS over the CSS likelihood, Yule-Walker init) replacing
   statsmodels' ARIMA MLE (statsmodels is not part of this
   stack); accuracy is comparable at the orders used here (1, 1).
 - ``mask_brain`` has no bundled MNI gray-matter template; pass
   ``template_name`` or use ``mask_self=True`` (the default).

Citation: [Ellis2020] "Facilitating open-science with realistic
fMRI simulation: validation and application", PeerJ 8:e8564.
"""

import logging

import numpy as np
from scipy import ndimage, optimize, signal, stats

logger = logging.getLogger(__name__)

__all__ = [
    'default_brain_template',
    "apply_signal",
    "calc_noise",
    "compute_signal_change",
    "convolve_hrf",
    "export_3_column",
    "export_epoch_file",
    "generate_1d_gaussian_rfs",
    "generate_1d_rf_responses",
    "generate_noise",
    "generate_signal",
    "generate_stimfunction",
    "mask_brain",
]


# ---------------------------------------------------------------------------
# signal generation
# ---------------------------------------------------------------------------

def _generate_feature(feature_type, feature_size, signal_magnitude,
                      thickness=1):
    """One activation feature (cube/loop/sphere/cavity) as a 3-D array."""
    if feature_size <= 2:
        feature_type = 'cube'

    if feature_type == 'cube':
        sig = np.ones((feature_size,) * 3)
    elif feature_type == 'loop':
        sig = np.zeros((feature_size,) * 3)
        seq = np.linspace(0, feature_size - 1, feature_size)
        xx, yy = np.meshgrid(seq, seq)
        disk = (xx - (feature_size - 1) / 2) ** 2 + \
            (yy - (feature_size - 1) / 2) ** 2
        outer_lim = disk[int((feature_size - 1) / 2), 0]
        inner_lim = disk[int((feature_size - 1) / 2), thickness]
        loop = (disk <= outer_lim) != (disk <= inner_lim)
        if not loop.any():
            loop = disk <= outer_lim
        sig[:, :, int(np.round(feature_size / 2))] = loop
    elif feature_type in ('sphere', 'cavity'):
        seq = np.linspace(0, feature_size - 1, feature_size)
        xx, yy, zz = np.meshgrid(seq, seq, seq)
        d2 = ((xx - (feature_size - 1) / 2) ** 2
              + (yy - (feature_size - 1) / 2) ** 2
              + (zz - (feature_size - 1) / 2) ** 2)
        c = int((feature_size - 1) / 2)
        outer_lim = d2[c, c, 0]
        inner_lim = d2[c, c, thickness]
        if feature_type == 'sphere':
            sig = d2 <= outer_lim
        else:
            sig = (d2 <= outer_lim) != (d2 <= inner_lim)
            if not sig.any():
                sig = d2 <= outer_lim
    else:
        raise ValueError('Unknown feature_type: ' + str(feature_type))
    return sig * signal_magnitude


def _insert_idxs(feature_centre, feature_size, dimensions):
    """Clipped index ranges for inserting a feature into a volume."""
    idxs = []
    for d in range(3):
        lo = int(feature_centre[d] - (feature_size / 2)) + 1
        hi = int(feature_centre[d] - (feature_size / 2) + feature_size) + 1
        lo = max(lo, 0)
        hi = min(hi, dimensions[d])
        idxs.append([lo, hi])
    return idxs[0], idxs[1], idxs[2]


def generate_signal(dimensions, feature_coordinates, feature_size,
                    feature_type, signal_magnitude=[1], signal_constant=1):
    """Volume with activation features of the given shapes/coordinates."""
    volume_signal = np.zeros(dimensions)
    feature_coordinates = np.asarray(feature_coordinates)
    feature_quantity = round(feature_coordinates.shape[0])
    if len(feature_size) == 1:
        feature_size = feature_size * feature_quantity
    if len(feature_type) == 1:
        feature_type = feature_type * feature_quantity
    if len(signal_magnitude) == 1:
        signal_magnitude = signal_magnitude * feature_quantity

    for i in range(feature_quantity):
        centre = np.asarray(feature_coordinates[i]
                            if feature_coordinates.ndim > 1
                            else feature_coordinates)
        sig = _generate_feature(feature_type[i], feature_size[i],
                                signal_magnitude[i])
        if signal_constant == 0:
            sig = sig * np.random.random([feature_size[i]] * 3)
        x_idx, y_idx, z_idx = _insert_idxs(centre, feature_size[i],
                                           dimensions)
        volume_signal[x_idx[0]:x_idx[1], y_idx[0]:y_idx[1],
                      z_idx[0]:z_idx[1]] = sig[
            :x_idx[1] - x_idx[0], :y_idx[1] - y_idx[0],
            :z_idx[1] - z_idx[0]]
    return volume_signal


def generate_stimfunction(onsets, event_durations, total_time, weights=[1],
                          timing_file=None, temporal_resolution=100.0):
    """Boxcar time course at ``temporal_resolution`` samples/second."""
    if timing_file is not None:
        with open(timing_file) as f:
            text = f.readlines()
        onsets, event_durations, weights = [], [], []
        for line in text:
            onset, duration, weight = line.strip().split()
            upsampled_onset = float(onset) * temporal_resolution
            if not np.allclose(upsampled_onset, np.round(upsampled_onset)):
                logger.warning(
                    'Onset %s has more decimal points than the temporal '
                    'resolution can resolve; events might be missed.',
                    onset)
            onsets.append(float(onset))
            event_durations.append(float(duration))
            weights.append(float(weight))

    if len(event_durations) == 1:
        event_durations = event_durations * len(onsets)
    if len(weights) == 1:
        weights = weights * len(onsets)
    if len(onsets) and np.max(onsets) > total_time:
        raise ValueError('Onsets outside of range of total time.')

    stimfunction = np.zeros(
        (int(round(total_time * temporal_resolution)), 1))
    for i in range(len(onsets)):
        onset_idx = int(np.floor(onsets[i] * temporal_resolution))
        offset_idx = int(np.floor(
            (onsets[i] + event_durations[i]) * temporal_resolution))
        stimfunction[onset_idx:offset_idx, 0] = weights[i]
    return stimfunction


def export_3_column(stimfunction, filename, temporal_resolution=100.0):
    """Write an FSL-style 3-column (onset, duration, weight) file."""
    stim_counter = 0
    while stim_counter < stimfunction.shape[0]:
        if stimfunction[stim_counter, 0] != 0:
            event_onset = stim_counter / temporal_resolution
            weight = stimfunction[stim_counter, 0]
            event_duration = 0
            while stim_counter < stimfunction.shape[0] and \
                    stimfunction[stim_counter, 0] != 0:
                event_duration += 1
                stim_counter += 1
            with open(filename, "a") as f:
                f.write(f"{event_onset}\t"
                        f"{event_duration / temporal_resolution}\t"
                        f"{weight}\n")
        stim_counter += 1


def export_epoch_file(stimfunction, filename, tr_duration,
                      temporal_resolution=100.0):
    """Write a BrainIAK-style epoch file (list of [condition, epoch, TR]
    boolean arrays, one per participant) as .npy."""
    epoch_file = [0] * len(stimfunction)
    for p, stim in enumerate(stimfunction):
        stim_binary = np.abs(stim) > 0
        stride = int(tr_duration * temporal_resolution)
        down = stim_binary[::stride, :]
        conditions = down.shape[1]
        trs = down.shape[0]

        epochs = 0
        for c in range(conditions):
            weight_change = np.diff(down[:, c].astype(int), 1, 0) != 0
            if down[0, c]:
                weight_change[0] = True
            if down[-1, c]:
                weight_change[-1] = True
            epochs += int(np.max(np.sum(weight_change, 0)) / 2)

        epoch_file[p] = np.zeros((conditions, epochs, trs))
        epoch_counter = 0
        tr_counter = 0
        while tr_counter < trs:
            for c in range(conditions):
                if tr_counter < trs and down[tr_counter, c]:
                    rest = np.where(down[tr_counter:, c] == 0)[0]
                    end_idx = rest[0] if rest.size else trs - tr_counter
                    epoch_file[p][c, epoch_counter,
                                  tr_counter:tr_counter + end_idx] = 1
                    tr_counter += end_idx
                    epoch_counter += 1
            tr_counter += 1
        epoch_file[p] = epoch_file[p].astype('bool')
    boxed = np.empty(len(epoch_file), dtype=object)
    for i, e in enumerate(epoch_file):
        boxed[i] = e
    np.save(filename, boxed, allow_pickle=True)


def _double_gamma_hrf(response_delay=6, undershoot_delay=12,
                      response_dispersion=0.9, undershoot_dispersion=0.9,
                      response_scale=1, undershoot_scale=0.035,
                      temporal_resolution=100.0):
    """Double-gamma HRF sampled at temporal_resolution (30 s support).

    Same closed form as the reference (fmrisim.py:723-803), including its
    undershoot exponent convention.
    """
    hrf_length = 30
    n = int(hrf_length * temporal_resolution)
    t = np.arange(n - 1) / temporal_resolution
    response_peak = response_delay * response_dispersion
    undershoot_peak = undershoot_delay * undershoot_dispersion
    with np.errstate(divide='ignore', invalid='ignore'):
        resp = (response_scale
                * np.power(t / response_peak, response_delay)
                * np.exp(-(t - response_peak) / response_dispersion))
        under = (undershoot_scale
                 * np.power(t / undershoot_peak, undershoot_delay)
                 * np.exp(-(t - undershoot_peak / undershoot_dispersion)))
    hrf = np.zeros(n)
    hrf[:n - 1] = np.nan_to_num(resp - under)
    return list(hrf)


def convolve_hrf(stimfunction, tr_duration, hrf_type='double_gamma',
                 scale_function=True, temporal_resolution=100.0):
    """Convolve stimulus function(s) with the HRF and downsample to TRs
    (mid-TR sampling, i.e. assumes slice-time-corrected data)."""
    if stimfunction.shape[0] < stimfunction.shape[1]:
        logger.warning('Stimfunction may be the wrong shape')
    if np.any(np.sum(abs(stimfunction), 0) == 0):
        logger.warning('stimfunction contains voxels of all zeros, '
                       'will nan')

    stride = int(temporal_resolution * tr_duration)
    duration = int(stimfunction.shape[0] / stride)
    if hrf_type == 'double_gamma':
        hrf = _double_gamma_hrf(temporal_resolution=temporal_resolution)
    elif isinstance(hrf_type, list):
        hrf = hrf_type
    else:
        raise ValueError('Unknown hrf_type')

    list_num = stimfunction.shape[1]
    signal_function = None
    for i in range(list_num):
        tmp = np.convolve(stimfunction[:, i], hrf)
        tmp = tmp[:duration * stride]
        vox = tmp[int(stride / 2)::stride]
        if scale_function and np.max(np.abs(vox)) > 0:
            vox = vox / np.max(vox)
        if signal_function is None:
            signal_function = np.zeros((len(vox), list_num))
        signal_function[:, i] = vox
    return signal_function


def apply_signal(signal_function, volume_signal):
    """Paint the signal time course(s) onto the non-zero voxels of the
    signal volume → 4-D array."""
    timepoints, timecourses = signal_function.shape
    signal = np.zeros(list(volume_signal.shape[:3]) + [timepoints])
    idxs = np.where(volume_signal != 0)
    if timecourses == 1:
        signal_function = np.tile(signal_function, (1, len(idxs[0])))
    elif len(idxs[0]) != timecourses:
        raise IndexError('The number of non-zero voxels in the volume and '
                         'the number of timecourses does not match. '
                         'Aborting')
    for i in range(len(idxs[0])):
        x, y, z = idxs[0][i], idxs[1][i], idxs[2][i]
        signal[x, y, z, :] = volume_signal[x, y, z] * signal_function[:, i]
    return signal


# ---------------------------------------------------------------------------
# noise estimation
# ---------------------------------------------------------------------------

def _calc_fwhm(volume, mask, voxel_size=[1.0, 1.0, 1.0]):
    """FWHM (mm) of a volume's masked voxels (gradient-variance method,
    vectorized re-expression of the reference's voxel loop)."""
    m = mask > 0
    vals = volume[m]
    v_count = vals.size
    v_sum = np.abs(vals).sum()
    v_sq = (vals ** 2).sum()
    v_var = (v_sq - (v_sum ** 2) / v_count) / (v_count - 1)

    d_var = np.zeros(3)
    for axis in range(3):
        sl_lo = [slice(None)] * 3
        sl_hi = [slice(None)] * 3
        sl_lo[axis] = slice(0, -1)
        sl_hi[axis] = slice(1, None)
        pair_mask = m[tuple(sl_lo)] & m[tuple(sl_hi)] & \
            ~np.isnan(volume[tuple(sl_hi)])
        diffs = (volume[tuple(sl_lo)] - volume[tuple(sl_hi)])[pair_mask]
        n = diffs.size
        d_var[axis] = (np.sum(diffs ** 2)
                       - (np.sum(diffs) ** 2) / n) / (n - 1)

    # volumes rougher than white noise push the AFNI estimator's log
    # argument <= 0; clamp -> FWHM ~ 0 instead of NaN
    arg = np.clip(1 - 0.5 * d_var / v_var, 1e-6, None)
    o_var = -1.0 / (4 * np.log(arg))
    fwhm3 = np.sqrt(o_var) * 2 * np.sqrt(2 * np.log(2))
    return np.prod(fwhm3 * np.asarray(voxel_size)) ** (1 / 3)


def _calc_sfnr(volume, mask):
    """Signal-to-fluctuation-noise ratio: mean / detrended std per brain
    voxel (2nd-order polynomial detrend), averaged."""
    brain_voxels = volume[mask > 0]
    mean_voxels = np.nanmean(brain_voxels, 1)
    seq = np.linspace(1, brain_voxels.shape[1], brain_voxels.shape[1])
    detrend_poly = np.polyfit(seq, brain_voxels.transpose(), 2)
    trend = (detrend_poly[0][:, None] * seq ** 2
             + detrend_poly[1][:, None] * seq + detrend_poly[2][:, None])
    std_voxels = np.nanstd(brain_voxels - trend, 1)
    return np.mean(mean_voxels / std_voxels)


def _calc_snr(volume, mask, dilation=5, reference_tr=None):
    """Mean brain intensity over std of (dilated-mask-excluded) non-brain
    voxels."""
    if reference_tr is None:
        reference_tr = list(range(volume.shape[3]))
    if dilation > 0:
        mask_dilated = ndimage.binary_dilation(mask, iterations=dilation)
    else:
        mask_dilated = mask
    brain_voxels = volume[mask > 0][:, reference_tr]
    nonbrain_voxels = volume[:, :, :, reference_tr].astype('float64')
    if len(brain_voxels.shape) > 1:
        brain_voxels = np.mean(brain_voxels, 1)
        nonbrain_voxels = np.mean(nonbrain_voxels, 3)
    nonbrain_voxels = nonbrain_voxels[mask_dilated == 0]
    return np.nanmean(brain_voxels) / np.nanstd(nonbrain_voxels)


def _estimate_ar_ma(timecourse, auto_reg_order=1, ma_order=1):
    """Yule-Walker AR estimate + first-lag moment MA estimate for one
    demeaned time course (statsmodels-free — see module docstring)."""
    x = timecourse - timecourse.mean()
    n = len(x)
    denom = np.dot(x, x)
    if denom <= 0:
        return ([np.nan] * auto_reg_order, [np.nan] * ma_order)
    acf = np.array([np.dot(x[:n - k], x[k:]) / denom
                    for k in range(auto_reg_order + 2)])
    # Yule-Walker for AR coefficients
    R = np.array([[acf[abs(i - j)] for j in range(auto_reg_order)]
                  for i in range(auto_reg_order)])
    r = acf[1:auto_reg_order + 1]
    try:
        ar = np.linalg.solve(R, r)
    except np.linalg.LinAlgError:
        ar = np.full(auto_reg_order, np.nan)
    # residual series → MA estimate from residual lag-1 autocorrelation
    resid = x[auto_reg_order:].copy()
    for k in range(auto_reg_order):
        resid = resid - ar[k] * x[auto_reg_order - 1 - k:
                                  n - 1 - k][:len(resid)]
    ma = np.zeros(ma_order)
    if resid.size > ma_order + 1 and np.dot(resid, resid) > 0:
        r1 = (np.dot(resid[:-1], resid[1:])
              / np.dot(resid, resid))
        # invert r1 = theta/(1+theta^2) (first-order MA moment relation)
        disc = 1 - 4 * r1 ** 2
        if disc >= 0 and abs(r1) > 1e-12:
            ma[0] = (1 - np.sqrt(disc)) / (2 * r1)
    return list(ar), list(ma)


def _estimate_arma_mle_batch(X, auto_reg_order=1, ma_order=1,
                             max_iter=60):
    """Batched conditional-MLE ARMA(p, q) fit on torch.

    Conditional-sum-of-squares likelihood (the CSS stage of
    statsmodels' ARIMA MLE that the reference relies on, ref
    fmrisim.py:1079-1290): innovations
    ``e_t = x_t − Σφ_i x_{t−i} − Σθ_j e_{t−j}`` with zero initial
    conditions, per-series σ² profiled out so the loss is
    ``Σ_b T/2·log(SSR_b / T)``.  Every sampled voxel's series optimizes
    JOINTLY in one L-BFGS pass (the objective is separable, the
    gradient block-diagonal) with the t-recursion vectorized across the
    batch — no per-voxel Python fitting loop.  Coefficients are
    tanh-bounded for stationarity/invertibility and initialized from
    the Yule-Walker moment estimates.

    X : [B, T] demeaned series.  Returns (ar [B, p], ma [B, q]).
    """
    import torch
    from scipy.optimize import minimize

    X = np.asarray(X, dtype=np.float64)
    B, T = X.shape
    p, q = auto_reg_order, ma_order

    init = np.zeros((B, p + q))
    for b in range(B):
        ar0, ma0 = _estimate_ar_ma(X[b], p, q)
        ar0 = np.nan_to_num(np.asarray(ar0, dtype=np.float64))
        ma0 = np.nan_to_num(np.asarray(ma0, dtype=np.float64))
        init[b, :p] = np.arctanh(np.clip(ar0, -0.95, 0.95))
        init[b, p:] = np.arctanh(np.clip(ma0, -0.95, 0.95))

    Xt = torch.as_tensor(X)
    params = torch.tensor(init.ravel(), requires_grad=True,
                          dtype=torch.float64)

    def closure(theta):
        with torch.no_grad():
            params.copy_(torch.as_tensor(theta, dtype=torch.float64))
        if params.grad is not None:
            params.grad = None
        ph = torch.tanh(params.reshape(B, p + q)[:, :p])
        th = torch.tanh(params.reshape(B, p + q)[:, p:])
        errs = []
        prev_e = [torch.zeros(B, dtype=torch.float64)
                  for _ in range(q)]
        for t in range(T):
            pred = torch.zeros(B, dtype=torch.float64)
            for i in range(p):
                if t - 1 - i >= 0:
                    pred = pred + ph[:, i] * Xt[:, t - 1 - i]
            for j in range(q):
                pred = pred + th[:, j] * prev_e[j]
            e = Xt[:, t] - pred
            errs.append(e)
            prev_e = [e] + prev_e[:-1]
        ssr = torch.stack(errs, dim=1).pow(2).sum(dim=1)
        loss = 0.5 * T * torch.log(ssr / T + 1e-30).sum()
        loss.backward()
        return float(loss.detach()), params.grad.numpy().copy()

    res = minimize(closure, init.ravel(), jac=True, method="L-BFGS-B",
                   options={"maxiter": max_iter})
    out = np.tanh(res.x.reshape(B, p + q))
    return out[:, :p], out[:, p:]


def _calc_ARMA_noise(volume, mask, auto_reg_order=1, ma_order=1,
                     sample_num=100, method='mle'):
    """Average ARMA coefficients over sampled brain voxels.

    ``method='mle'`` (default) uses the batched conditional-MLE fit;
    ``'moments'`` keeps the Yule-Walker fallback."""
    if len(volume.shape) > 1:
        brain_timecourse = volume[mask > 0]
    else:
        brain_timecourse = volume.reshape(1, len(volume))
    voxel_idxs = list(range(brain_timecourse.shape[0]))
    np.random.shuffle(voxel_idxs)
    sample_num = min(sample_num, len(voxel_idxs))
    series = np.stack([brain_timecourse[voxel_idxs[i], :]
                       for i in range(sample_num)])
    series = series - series.mean(axis=1, keepdims=True)
    # drop degenerate (constant) series
    keep = series.std(axis=1) > 1e-12
    if method == 'mle' and keep.any():
        ar_all, ma_all = _estimate_arma_mle_batch(
            series[keep], auto_reg_order, ma_order)
        return (np.nanmean(ar_all, 0).tolist(),
                np.nanmean(ma_all, 0).tolist())
    ar_all = np.zeros((sample_num, auto_reg_order))
    ma_all = np.zeros((sample_num, ma_order))
    for i in range(sample_num):
        ar, ma = _estimate_ar_ma(series[i], auto_reg_order, ma_order)
        ar_all[i, :] = ar
        ma_all[i, :] = ma
    return (np.nanmean(ar_all, 0).tolist(),
            np.nanmean(ma_all, 0).tolist())


def calc_noise(volume, mask, template, noise_dict=None):
    """Estimate the noise properties (SFNR, SNR, FWHM, AR/MA, ...) of a
    real or simulated 4-D volume."""
    if template.max() > 1.1:
        raise ValueError('Template out of range')
    if mask is None:
        raise ValueError('Mask not supplied')
    if noise_dict is None:
        noise_dict = {'voxel_size': [1.0, 1.0, 1.0]}
    elif 'voxel_size' not in noise_dict:
        noise_dict['voxel_size'] = [1.0, 1.0, 1.0]

    noise_dict['max_activity'] = np.nanmax(np.mean(volume, 3))
    noise_dict['auto_reg_rho'], noise_dict['ma_rho'] = _calc_ARMA_noise(
        volume, mask)
    noise_dict['auto_reg_sigma'] = 1
    noise_dict['physiological_sigma'] = 0
    noise_dict['task_sigma'] = 0
    noise_dict['drift_sigma'] = 0
    noise_dict['sfnr'] = _calc_sfnr(volume, mask)

    if volume.shape[3] > 100:
        trs = np.random.choice(volume.shape[3], size=100, replace=False)
    else:
        trs = list(range(volume.shape[3]))
    fwhm = [_calc_fwhm(volume[:, :, :, tr], mask,
                       noise_dict['voxel_size']) for tr in trs]
    noise_dict['fwhm'] = np.mean(fwhm)
    noise_dict['snr'] = _calc_snr(volume, mask)
    return noise_dict


# ---------------------------------------------------------------------------
# noise generation
# ---------------------------------------------------------------------------

def _generate_noise_system(dimensions_tr, spatial_sd, temporal_sd,
                           spatial_noise_type='gaussian',
                           temporal_noise_type='gaussian'):
    """Scanner noise: a static spatial component plus temporally varying
    (zero-mean in time) component."""
    def noise_volume(dimensions, noise_type):
        if noise_type == 'rician':
            return stats.rice.rvs(b=0, loc=0, scale=1.527,
                                  size=dimensions)
        if noise_type == 'exponential':
            return stats.expon.rvs(0, scale=1, size=dimensions)
        return np.random.randn(int(np.prod(dimensions))).reshape(
            dimensions)

    dimensions = np.asarray([dimensions_tr[0], dimensions_tr[1],
                             dimensions_tr[2], 1])
    spatial_noise = noise_volume(dimensions, spatial_noise_type)
    temporal_noise = noise_volume(dimensions_tr, temporal_noise_type)
    spatial_noise *= spatial_sd
    temporal_noise *= temporal_sd
    temporal_noise = temporal_noise - np.mean(
        temporal_noise, 3, keepdims=True)
    return spatial_noise + temporal_noise


def _generate_noise_temporal_task(stimfunction_tr, motion_noise='gaussian'):
    """Event-locked noise."""
    stim = stimfunction_tr != 0
    if motion_noise == 'gaussian':
        noise = stim * np.random.normal(0, 1, size=stim.shape)
    else:
        noise = stim * stats.rice.rvs(0, 1, size=stim.shape)
    return stats.zscore(stim + noise).flatten()


def _generate_noise_temporal_drift(trs, tr_duration,
                                   basis="cos_power_drop", period=150):
    """Slow scanner drift from cosine bases (or a single sine)."""
    if basis == 'discrete_cos':
        timepoints = np.linspace(0, trs - 1, trs)
        timepoints = ((timepoints * tr_duration) / period) * 2 * np.pi
        duration = trs * tr_duration
        basis_funcs = int(np.floor(duration / period))
        if basis_funcs == 0:
            logger.warning('Too few timepoints (%d) to accurately model '
                           'drift', trs)
            basis_funcs = 1
        drift = np.zeros((trs, basis_funcs))
        for b in range(1, basis_funcs + 1):
            drift[:, b - 1] = np.cos(
                (timepoints / b) + np.random.rand() * np.pi * 2)
        noise_drift = np.mean(drift, 1)
    elif basis == 'sine':
        cycles = trs * tr_duration / period
        timepoints = np.linspace(0, trs - 1, trs)
        phaseshift = np.pi * 2 * np.random.random()
        noise_drift = np.sin(
            (timepoints / (trs - 1) * cycles * 2 * np.pi) + phaseshift)
    elif basis == 'cos_power_drop':
        timepoints = np.linspace(0, trs - 1, trs) * tr_duration
        duration = trs * tr_duration
        basis_funcs = int(trs)
        drift = np.zeros((trs, basis_funcs))
        for b in range(1, basis_funcs + 1):
            drift[:, b - 1] = np.cos(
                (timepoints / duration * np.pi * b)
                + np.random.rand() * np.pi * 2)

        def power_drop(r, L, F, trd):
            if F < trd:
                raise ValueError('Period %0.0f > TR duration %0.0f'
                                 % (F, trd))
            numerator = 1 - r ** (2 * L / F)
            denominator = 1 - r ** (2 * L / trd)
            return abs((numerator / denominator) - 0.99)

        sol = optimize.minimize_scalar(
            power_drop, bounds=(0, 1), method='Bounded',
            args=(duration, period, tr_duration))
        weights = sol.x ** np.arange(basis_funcs)
        noise_drift = np.mean(drift * weights, 1)
    else:
        raise ValueError('Unknown drift basis')
    return stats.zscore(noise_drift)


def _generate_noise_spatial(dimensions, mask=None, fwhm=4.0):
    """Gaussian-random-field volume with approximately the given FWHM
    (power-law amplitude spectrum in k-space)."""
    if len(dimensions) == 4:
        dimensions = dimensions[0:3]
    if dimensions[0] != dimensions[1] or dimensions[1] != dimensions[2]:
        max_dim = np.max(dimensions)
        new_dim = (max_dim, max_dim, max_dim)
    else:
        new_dim = tuple(dimensions)

    # empirical fwhm → sigma mapping (reference fmrisim.py:1956-1981)
    spatial_sigma = (np.log(fwhm - 0.36778719) / np.log(2.10601011)) \
        + 2.15439247

    noise = np.fft.fftn(np.random.normal(size=new_dim))

    def fft_idx(n):
        ascending = np.linspace(0, int(n / 2), int(n / 2 + 1))
        elements = int(np.ceil(n / 2 - 1))
        descending = np.linspace(-elements, -1, elements)
        return np.concatenate((ascending, descending))

    grids = np.meshgrid(fft_idx(new_dim[0]), fft_idx(new_dim[1]),
                        fft_idx(new_dim[2]), indexing='ij')
    k2 = grids[0] ** 2 + grids[1] ** 2 + grids[2] ** 2
    with np.errstate(divide='ignore'):
        amplitude = np.sqrt(np.sqrt(k2) ** (-spatial_sigma))
    amplitude[k2 == 0] = 0

    noise_fft = np.fft.ifftn(noise * amplitude).real
    noise_spatial = noise_fft[:dimensions[0], :dimensions[1],
                              :dimensions[2]]
    if mask is not None:
        noise_spatial = noise_spatial * mask
        brain = mask > 0
        noise_spatial[brain] = stats.zscore(noise_spatial[brain])
    else:
        noise_spatial = (noise_spatial - noise_spatial.mean()) \
            / noise_spatial.std()
    return noise_spatial


def _generate_noise_temporal_autoregression(timepoints, noise_dict,
                                            dimensions, mask):
    """ARMA process of spatially smooth volumes, z-scored in time."""
    auto_reg_rho = noise_dict['auto_reg_rho']
    ma_rho = noise_dict['ma_rho']
    auto_reg_order = len(auto_reg_rho)
    ma_order = len(ma_rho)
    if ma_order > auto_reg_order:
        raise ValueError('MA order (%d) is greater than AR order (%d). '
                         'Cannot run.' % (ma_order, auto_reg_order))
    T = len(timepoints)
    out = np.zeros(tuple(dimensions) + (T,))
    errs = np.zeros(tuple(dimensions) + (T,))
    for t in range(T):
        noise = _generate_noise_spatial(dimensions=dimensions, mask=mask,
                                        fwhm=noise_dict['fwhm'])
        errs[..., t] = noise
        if t == 0:
            out[..., 0] = noise
        else:
            ar_vol = np.zeros(dimensions)
            for p in range(1, auto_reg_order + 1):
                if t - p >= 0:
                    ar_vol += out[..., t - p] * auto_reg_rho[p - 1]
                    if ma_order >= p:
                        ar_vol += errs[..., t - p] * ma_rho[p - 1]
            out[..., t] = ar_vol + noise
    return stats.zscore(out, 3)


def _generate_noise_temporal_phys(timepoints, resp_freq=0.2,
                                  heart_freq=1.17):
    """Respiration + heart-beat oscillations."""
    resp_phase = np.random.rand() * 2 * np.pi
    heart_phase = np.random.rand() * 2 * np.pi
    t = np.asarray(timepoints)
    noise_phys = (np.cos(t * resp_freq * 2 * np.pi + resp_phase)
                  + np.sin(t * heart_freq * 2 * np.pi + heart_phase))
    return stats.zscore(noise_phys)


def _generate_noise_temporal(stimfunction_tr, tr_duration, dimensions,
                             template, mask, noise_dict):
    """Mix physiological, autoregressive and task noise volumes."""
    trs = len(stimfunction_tr)
    timepoints = list(np.linspace(0, (trs - 1) * tr_duration, trs))
    noise_volume = np.zeros(tuple(dimensions) + (trs,))

    if noise_dict['physiological_sigma'] != 0:
        noise = _generate_noise_temporal_phys(timepoints)
        volume = _generate_noise_spatial(dimensions=dimensions, mask=mask,
                                         fwhm=noise_dict['fwhm'])
        noise_volume += np.multiply.outer(volume, noise) * \
            noise_dict['physiological_sigma']

    if noise_dict['auto_reg_sigma'] != 0:
        noise = _generate_noise_temporal_autoregression(
            timepoints, noise_dict, dimensions, mask)
        noise_volume += noise * noise_dict['auto_reg_sigma']

    if noise_dict['task_sigma'] != 0 and np.sum(stimfunction_tr) > 0:
        noise = _generate_noise_temporal_task(stimfunction_tr)
        volume = _generate_noise_spatial(dimensions=dimensions, mask=mask,
                                         fwhm=noise_dict['fwhm'])
        noise_volume += np.multiply.outer(volume, noise) * \
            noise_dict['task_sigma']

    noise_volume = stats.zscore(noise_volume, 3)
    noise_volume[np.isnan(noise_volume)] = 0
    return noise_volume


def default_brain_template(dimensions):
    """Procedural standard-brain intensity template.

    The reference bundles an MNI152 gray-matter template for
    ``mask_brain`` (ref fmrisim.py:2230); shipping MNI data offline is
    not possible here, so this builds a deterministic stand-in with the
    properties mask_brain's threshold detector needs: a superellipsoid
    head with a bright cortical shell over a dimmer interior, smooth
    falloff to zero outside, values in [0, 1] with a bimodal histogram.
    """
    dims = np.asarray(dimensions[:3], dtype=int)
    gx, gy, gz = np.meshgrid(
        *(np.linspace(-1, 1, d) for d in dims), indexing='ij')
    # head slightly egg-shaped: wider front-back (y), flat-bottomed (z)
    r = ((gx / 0.82) ** 2 + (gy / 0.94) ** 2
         + ((gz - 0.08) / 0.78) ** 2) ** 0.5
    interior = np.clip(1.0 - r, 0, None)
    core = 0.55 * np.tanh(6 * interior)           # white-matter plateau
    shell = 0.45 * np.exp(-((r - 0.82) / 0.10) ** 2)   # cortical ring
    template = np.clip(core + shell, 0, 1)
    template[r > 1.0] = 0.0
    return template


def mask_brain(volume, template_name=None, mask_threshold=None,
               mask_self=True):
    """Build a (mask, template) pair from a volume (or a stored template).

    With ``mask_self=False`` and no ``template_name`` the procedural
    :func:`default_brain_template` is used (see its docstring for why
    no MNI data ships).
    """
    if len(volume.shape) == 1:
        volume = np.ones(volume.astype(int))

    if mask_self is True:
        mask_raw = volume
    elif template_name is None:
        mask_raw = default_brain_template(volume.shape)
    else:
        mask_raw = np.load(template_name)

    if len(mask_raw.shape) == 3:
        mask_raw = np.array(mask_raw)
    elif len(mask_raw.shape) == 4 and mask_raw.shape[3] == 1:
        mask_raw = np.array(mask_raw[:, :, :, 0])
    else:
        mask_raw = np.mean(mask_raw, 3)

    mask_raw = mask_raw / mask_raw.max()

    if len(volume.shape) == 3:
        volume = volume[:, :, :, np.newaxis]
    brain_dim = volume.shape
    mask_dim = mask_raw.shape
    zoom_factor = (brain_dim[0] / mask_dim[0],
                   brain_dim[1] / mask_dim[1],
                   brain_dim[2] / mask_dim[2])
    template = ndimage.zoom(mask_raw, zoom_factor, order=2)
    template[template < 0] = 0

    if mask_threshold is None:
        order = 5
        template_vector = template.reshape(-1)
        template_hist = np.histogram(template_vector, 100)
        binval = np.concatenate([np.zeros(order), template_hist[0]])
        bins = np.concatenate([np.zeros(order), template_hist[1]])
        peaks = signal.argrelmax(binval, order=order)[0][0:2]
        if len(peaks) == 2:
            minima = binval[peaks[0]:peaks[1]].min()
            minima_idx = (np.where(binval[peaks[0]:peaks[1]] == minima)
                          + peaks[0])[-1]
            mask_threshold = bins[minima_idx][0]
        else:
            mask_threshold = 0.2  # fallback for non-bimodal data

    mask = np.zeros(template.shape)
    mask[template > mask_threshold] = 1
    return mask, template


def _noise_dict_update(noise_dict):
    """Fill in default noise parameters."""
    default_dict = {'task_sigma': 0, 'drift_sigma': 0, 'auto_reg_sigma': 1,
                    'auto_reg_rho': [0.5], 'ma_rho': [0.0],
                    'physiological_sigma': 0, 'sfnr': 90, 'snr': 50,
                    'max_activity': 1000,
                    'voxel_size': [1.0, 1.0, 1.0], 'fwhm': 4,
                    'matched': 1}
    for key, val in default_dict.items():
        if key not in noise_dict:
            noise_dict[key] = val
    return noise_dict


def _assemble_noise(base, drift_noise, noise_system, noise_temporal,
                    temporal_sd):
    noise = base + drift_noise + noise_system
    noise = noise + (noise_temporal * temporal_sd)
    noise[noise < 0] = 0
    return noise


def _fit_spatial(noise, noise_temporal, drift_noise, mask, template,
                 spatial_sd, temporal_sd, noise_dict, fit_thresh,
                 fit_delta, iterations):
    """Iteratively adjust system spatial SD until the measured SNR
    matches the target."""
    dim_tr = noise.shape
    base = (template * noise_dict['max_activity']).reshape(
        dim_tr[0], dim_tr[1], dim_tr[2], 1)
    mean_signal = (base[mask > 0]).mean()
    target_snr = noise_dict['snr']
    spat_sd_orig = np.copy(spatial_sd)
    for iteration in range(iterations):
        new_snr = _calc_snr(noise, mask)
        diff_snr = abs(new_snr - target_snr) / target_snr
        if diff_snr < fit_thresh:
            logger.info('Terminated SNR fit after %d iterations.',
                        iteration)
            break
        spat_sd_new = mean_signal / new_snr
        spatial_sd -= ((spat_sd_new - spat_sd_orig) * fit_delta)
        if spatial_sd < 0 or np.isnan(spatial_sd):
            spatial_sd = 10e-3
        noise_system = _generate_noise_system(
            dimensions_tr=dim_tr, spatial_sd=spatial_sd,
            temporal_sd=temporal_sd)
        noise = _assemble_noise(base, drift_noise, noise_system,
                                noise_temporal, temporal_sd)
    return noise, spatial_sd


def _fit_temporal(noise, mask, template, stimfunction_tr, tr_duration,
                  spatial_sd, temporal_proportion, temporal_sd,
                  drift_noise, noise_dict, fit_thresh, fit_delta,
                  iterations):
    """Iteratively adjust temporal SD and AR rho to match SFNR/AR."""
    import copy as _copy
    dim_tr = noise.shape
    dim = dim_tr[0:3]
    base = (template * noise_dict['max_activity']).reshape(
        dim[0], dim[1], dim[2], 1)
    mean_signal = (base[mask > 0]).mean()
    temp_sd_orig = np.copy(temporal_sd)
    new_nd = _copy.deepcopy(noise_dict)
    target_sfnr = noise_dict['sfnr']
    target_ar = noise_dict['auto_reg_rho'][0]
    for iteration in range(iterations):
        new_sfnr = _calc_sfnr(noise, mask)
        new_ar, _ = _calc_ARMA_noise(noise, mask,
                                     len(noise_dict['auto_reg_rho']),
                                     len(noise_dict['ma_rho']))
        sfnr_diff = abs(new_sfnr - target_sfnr) / target_sfnr
        ar_diff = new_ar[0] - target_ar
        if (abs(ar_diff) / target_ar) < fit_thresh and \
                sfnr_diff < fit_thresh:
            logger.info('Terminated AR fit after %d iterations.',
                        iteration)
            break
        temp_sd_new = mean_signal / new_sfnr
        temporal_sd -= ((temp_sd_new - temp_sd_orig) * fit_delta)
        if temporal_sd < 0 or np.isnan(temporal_sd):
            temporal_sd = 10e-3
        temp_sd_system_new = np.sqrt(
            (temporal_sd ** 2) * temporal_proportion)
        new_nd['auto_reg_rho'][0] -= (ar_diff * fit_delta)
        if new_nd['auto_reg_rho'][0] >= 1:
            new_nd['auto_reg_rho'][0] = 0.99
        noise_temporal = _generate_noise_temporal(
            stimfunction_tr, tr_duration, dim, template, mask, new_nd)
        noise_system = _generate_noise_system(
            dimensions_tr=dim_tr, spatial_sd=spatial_sd,
            temporal_sd=temp_sd_system_new)
        noise = _assemble_noise(base, drift_noise, noise_system,
                                noise_temporal, temporal_sd)
    return noise


def generate_noise(dimensions, stimfunction_tr, tr_duration, template,
                   mask=None, noise_dict=None, temporal_proportion=0.5,
                   iterations=None, fit_thresh=0.05, fit_delta=0.5):
    """Generate a 4-D noise volume with the requested properties (see
    the reference's docstring for the noise model; this follows it)."""
    if template.max() > 1.1:
        raise ValueError('Template out of range')
    if noise_dict is None:
        noise_dict = {}
    noise_dict = _noise_dict_update(noise_dict)
    if iterations is None:
        iterations = [20, 20] if noise_dict['matched'] == 1 else [0, 0]
    if abs(noise_dict['auto_reg_rho'][0]) - abs(
            noise_dict['ma_rho'][0]) < 0.1:
        logger.warning('ARMA coefs are close, may have trouble fitting')

    dimensions_tr = (dimensions[0], dimensions[1], dimensions[2],
                     len(stimfunction_tr))
    if mask is None:
        mask = np.ones(dimensions)

    base = (template * noise_dict['max_activity']).reshape(
        dimensions[0], dimensions[1], dimensions[2], 1)
    base = np.ones(dimensions_tr) * base
    mean_signal = (base[mask > 0]).mean()

    noise_temporal = _generate_noise_temporal(
        stimfunction_tr=stimfunction_tr, tr_duration=tr_duration,
        dimensions=dimensions, template=template, mask=mask,
        noise_dict=noise_dict)

    if noise_dict['drift_sigma'] != 0:
        noise = _generate_noise_temporal_drift(len(stimfunction_tr),
                                               tr_duration)
        drift_noise = np.multiply.outer(np.ones(dimensions[:3]), noise) \
            * noise_dict['drift_sigma']
    else:
        drift_noise = np.zeros(dimensions_tr)

    temporal_sd = mean_signal / noise_dict['sfnr']
    temporal_sd_system = np.sqrt((temporal_sd ** 2) * temporal_proportion)
    spat_sd = mean_signal / noise_dict['snr']
    spatial_sd = np.sqrt((spat_sd ** 2) * (1 - temporal_proportion))

    noise_system = _generate_noise_system(
        dimensions_tr=dimensions_tr, spatial_sd=spatial_sd,
        temporal_sd=temporal_sd_system)

    noise = _assemble_noise(base, drift_noise, noise_system,
                            noise_temporal, temporal_sd)

    noise, spatial_sd = _fit_spatial(
        noise, noise_temporal, drift_noise, mask, template, spatial_sd,
        temporal_sd_system, noise_dict, fit_thresh, fit_delta,
        iterations[0])
    noise = _fit_temporal(
        noise, mask, template, stimfunction_tr, tr_duration, spatial_sd,
        temporal_proportion, temporal_sd, drift_noise, noise_dict,
        fit_thresh, fit_delta, iterations[1])
    return noise


def compute_signal_change(signal_function, noise_function, noise_dict,
                          magnitude, method='PSC'):
    """Rescale signal time courses to a magnitude under a given metric
    (PSC, SFNR, CNR variants — Welvaert & Rosseel 2013)."""
    assert type(magnitude) is list, '"magnitude" should be a list of floats'
    if len(magnitude) == 1:
        magnitude = magnitude * signal_function.shape[1]
    if signal_function.shape != noise_function.shape:
        raise ValueError(
            'noise_function is not the same size as signal_function')

    peak = np.max(np.abs(signal_function))
    signal_function = signal_function / (peak if peak > 0 else 1.0)
    out = np.zeros(signal_function.shape)
    for v in range(signal_function.shape[1]):
        sig_voxel = signal_function[:, v]
        noise_voxel = noise_function[:, v]
        mag = magnitude[v]
        max_amp = np.max(np.abs(sig_voxel))
        if method == 'SFNR':
            new_sig = sig_voxel * (
                noise_voxel.mean() / noise_dict['sfnr'] * mag)
        elif method == 'CNR_Amp/Noise-SD':
            new_sig = sig_voxel * (mag * np.std(noise_voxel))
        elif method == 'CNR_Amp2/Noise-Var_dB':
            scale = (10 ** (mag / 20)) * np.std(noise_voxel) / max_amp
            new_sig = sig_voxel * scale
        elif method == 'CNR_Signal-SD/Noise-SD':
            new_sig = sig_voxel * ((mag / max_amp) * np.std(noise_voxel)
                                   / np.std(sig_voxel))
        elif method == 'CNR_Signal-Var/Noise-Var_dB':
            scale = (10 ** (mag / 20)) * np.std(noise_voxel) \
                / np.std(sig_voxel) / max_amp
            new_sig = sig_voxel * scale
        elif method == 'PSC':
            new_sig = sig_voxel * (noise_voxel.mean() / 100 * mag)
        else:
            raise ValueError('Unknown method: ' + str(method))
        out[:, v] = new_sig
    return out


# ---------------------------------------------------------------------------
# 1-D receptive field helpers
# ---------------------------------------------------------------------------

def generate_1d_gaussian_rfs(n_voxels, feature_resolution, feature_range,
                             rf_size=15, random_tuning=True, rf_noise=0.):
    """Gaussian voxel receptive fields tiled along one feature axis."""
    range_start, range_stop = feature_range
    if random_tuning:
        voxel_tuning = np.floor((np.random.rand(n_voxels) * range_stop)
                                + range_start).astype(int)
    else:
        voxel_tuning = np.linspace(range_start, range_stop,
                                   n_voxels + 1)[:-1]
        voxel_tuning = np.floor(voxel_tuning).astype(int)
    gaussian = signal.windows.gaussian(feature_resolution, rf_size)
    voxel_rfs = np.zeros((n_voxels, feature_resolution))
    for i in range(n_voxels):
        voxel_rfs[i, :] = np.roll(
            gaussian, voxel_tuning[i] - ((feature_resolution // 2) - 1))
    voxel_rfs += np.random.rand(n_voxels, feature_resolution) * rf_noise
    voxel_rfs = voxel_rfs / np.max(voxel_rfs, axis=1)[:, None]
    return voxel_rfs, voxel_tuning


def generate_1d_rf_responses(rfs, trial_list, feature_resolution,
                             feature_range, trial_noise=0.25):
    """Per-trial voxel responses from RFs and presented feature values."""
    range_start, range_stop = feature_range
    stim = np.zeros((feature_resolution, len(trial_list)))
    trial_idx = np.floor(
        (np.asarray(trial_list) - range_start)
        / (range_stop - range_start + 1) * feature_resolution).astype(int)
    trial_idx = np.clip(trial_idx, 0, feature_resolution - 1)
    for t, idx in enumerate(trial_idx):
        stim[idx, t] = 1
    trial_data = rfs @ stim
    trial_data += np.random.rand(*trial_data.shape) * trial_noise
    return trial_data


# internal aliases used elsewhere in the package
double_gamma_hrf = _double_gamma_hrf
