"""Real-time fMRI stream simulator (API parity: ref
src/brainiak/utils/fmrisim_real_time_generator.py:349-637).

Writes one volume per TR to ``outputDir`` (``rt_###.npy``), optionally
paced in real time, with condition labels and a brain mask — the input
stream for real-time analysis pipelines.

Deviations: no bundled ROI/template NIfTIs — defaults are generated
synthetically (two spherical ROIs inside a smooth ellipsoid template).
``save_dicom=True`` writes real DICOM Part-10 files (``rt_###.dcm``,
Explicit-VR LE multi-frame, via the self-contained
``utils.dicom_minimal`` writer — pydicom is not installed here).
"""

import datetime
import logging
import os
import time
from pathlib import Path

import numpy as np

from . import fmrisim as sim

logger = logging.getLogger(__name__)

# reference-parity module attribute: the generator's start timestamp
# (the reference stamps DICOM acquisition times relative to it,
# ref fmrisim_real_time_generator.py:49)
script_datetime = datetime.datetime.now()

__all__ = ["generate_data"]

default_settings = {
    'ROI_A_file': None,
    'ROI_B_file': None,
    'template_path': None,
    'noise_dict_file': None,
    'numTRs': 200,
    'trDuration': 2,
    'isi': 6,
    'burn_in': 6,
    'event_duration': 10,
    'scale_percentage': 0.5,
    'multivariate_pattern': False,
    'different_ROIs': False,
    'save_dicom': False,
    'save_realtime': False,
    'dimensions': (24, 24, 16),
}


def _default_template(dimensions):
    """Smooth ellipsoid 'brain' template in [0, 1]."""
    grids = np.meshgrid(*[np.linspace(-1, 1, d) for d in dimensions],
                        indexing='ij')
    r2 = sum(g ** 2 for g in grids)
    template = np.clip(1.1 - r2, 0, 1)
    return template / template.max()


def _default_roi(dimensions, centre_frac, radius=2):
    """Small spherical ROI at a fractional position in the volume."""
    centre = [int(d * f) for d, f in zip(dimensions, centre_frac)]
    grids = np.meshgrid(*[np.arange(d) for d in dimensions],
                        indexing='ij')
    r2 = sum((g - c) ** 2 for g, c in zip(grids, centre))
    return (r2 <= radius ** 2).astype(float)


def _generate_ROIs(ROI, stimfunc, noise, scale_percentage, data_dict):
    """Evoked-response signal volume for one ROI, scaled to a percent
    signal change against the noise."""
    vx, vy, vz = np.nonzero(ROI == 1)
    voxels = len(vx)

    if data_dict['multivariate_pattern']:
        per_voxel = np.random.rand(1, voxels)      # random multivoxel map
    else:
        per_voxel = np.ones((1, voxels))           # uniform response
    weights = stimfunc * per_voxel

    signal_func = sim.convolve_hrf(
        stimfunction=weights,
        tr_duration=data_dict['trDuration'],
        temporal_resolution=1 / data_dict['trDuration'],
        scale_function=1)
    roi_noise = noise.astype('double')[vx, vy, vz, :].T
    sf_scaled = sim.compute_signal_change(
        signal_function=signal_func, noise_function=roi_noise,
        noise_dict=data_dict['noise_dict'],
        magnitude=[scale_percentage], method='PSC')
    return sim.apply_signal(sf_scaled, ROI)


def generate_data(outputDir, user_settings):
    """Generate and stream simulated fMRI volumes to ``outputDir``.

    See ``default_settings`` for parameters; same contract as the
    reference's generator.
    """
    data_dict = default_settings.copy()
    data_dict.update(user_settings)

    Path(outputDir).mkdir(parents=True, exist_ok=True)

    # template + ROIs (synthetic defaults when no files are given)
    if data_dict.get('template_path') is None:
        template = _default_template(data_dict['dimensions'])
    elif isinstance(data_dict['template_path'], str):
        template = np.load(data_dict['template_path'])
    else:
        template = data_dict['template_path']
    dimensions = np.array(template.shape[0:3])

    mask, template = sim.mask_brain(volume=template, mask_self=True)
    np.save(os.path.join(outputDir, 'mask.npy'), mask.astype(np.uint8))

    def _load_roi(key, default_frac):
        val = data_dict.get(key)
        if val is None:
            return _default_roi(dimensions, default_frac)
        if isinstance(val, str):
            return np.load(val)
        return val

    ROI_A = _load_roi('ROI_A_file', (0.35, 0.5, 0.5))
    ROI_B = _load_roi('ROI_B_file', (0.65, 0.5, 0.5))

    if data_dict.get('noise_dict_file') is None:
        noise_dict = {'matched': 0}
    elif isinstance(data_dict['noise_dict_file'], str):
        with open(data_dict['noise_dict_file']) as f:
            noise_dict = eval(f.read())  # same format as the reference
        noise_dict['matched'] = 0
    else:
        noise_dict = dict(data_dict['noise_dict_file'])
        noise_dict['matched'] = 0
    data_dict['noise_dict'] = sim._noise_dict_update(noise_dict)

    logger.info('Generating noise')
    temp_stimfunction = np.zeros((data_dict['numTRs'], 1))
    noise = sim.generate_noise(dimensions=dimensions,
                               stimfunction_tr=temp_stimfunction,
                               tr_duration=int(data_dict['trDuration']),
                               template=template, mask=mask,
                               noise_dict=noise_dict)

    # randomized A/B block design
    total_time = int(data_dict['numTRs'] * data_dict['trDuration'])
    slot = data_dict['event_duration'] + data_dict['isi']
    slot_starts = np.arange(data_dict['burn_in'],
                            total_time - data_dict['event_duration'],
                            slot)
    coin = np.random.randint(0, 2, size=len(slot_starts))
    onsets_A = [float(t) for t, c in zip(slot_starts, coin) if c == 1]
    onsets_B = [float(t) for t, c in zip(slot_starts, coin) if c == 0]

    def _stimfunction(onsets):
        return sim.generate_stimfunction(
            onsets=onsets,
            event_durations=[data_dict['event_duration']],
            total_time=total_time,
            temporal_resolution=1 / data_dict['trDuration'])

    stimfunc_A = _stimfunction(onsets_A)
    stimfunc_B = _stimfunction(onsets_B)
    np.save(os.path.join(outputDir, 'labels.npy'),
            stimfunc_A + (stimfunc_B * 2))

    signal_A = _generate_ROIs(ROI_A, stimfunc_A, noise,
                              data_dict['scale_percentage'], data_dict)
    if data_dict['different_ROIs'] is True:
        signal_B = _generate_ROIs(ROI_B, stimfunc_B, noise,
                                  data_dict['scale_percentage'],
                                  data_dict)
    elif data_dict['multivariate_pattern'] is False:
        signal_B = _generate_ROIs(ROI_A, stimfunc_B, noise,
                                  data_dict['scale_percentage'] * 0.5,
                                  data_dict)
    else:
        signal_B = _generate_ROIs(ROI_A, stimfunc_B, noise,
                                  data_dict['scale_percentage'],
                                  data_dict)
    signal = signal_A + signal_B

    logger.info('Generating TRs in real time')
    composed = noise + signal
    for idx in range(data_dict['numTRs']):
        start = time.time()
        brain = composed[..., idx]
        brain_int32 = np.nan_to_num(brain).astype(np.int32)
        if data_dict['save_dicom']:
            from .dicom_minimal import write_dicom
            output_file = os.path.join(
                outputDir, 'rt_' + format(idx, '03d') + '.dcm')
            write_dicom(output_file, np.clip(brain_int32, 0, 65535),
                        instance_number=idx + 1,
                        tr_seconds=data_dict['trDuration'])
        else:
            output_file = os.path.join(
                outputDir, 'rt_' + format(idx, '03d') + '.npy')
            np.save(output_file, brain_int32)
        if data_dict['save_realtime']:
            elapsed = time.time() - start
            remaining = data_dict['trDuration'] - elapsed
            if remaining > 0:
                time.sleep(remaining)
    logger.info('Generated %d volumes in %s', data_dict['numTRs'],
                outputDir)
