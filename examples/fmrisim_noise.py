#!/usr/bin/env python3
"""fmrisim: generate task signal + realistic noise, then verify the
signal is recoverable (the reference's fmrisim example)."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.utils import fmrisim


def main():
    rng = np.random.RandomState(0)
    dim = (10, 10, 10)
    tr, trs = 2.0, 100

    onsets = np.arange(10, 180, 30.0)
    stim = fmrisim.generate_stimfunction(
        onsets=onsets, event_durations=[4.0], total_time=trs * tr)
    signal_func = fmrisim.convolve_hrf(stim, tr_duration=tr)

    coords = np.array([[5, 5, 5]])
    vol = fmrisim.generate_signal(
        dimensions=np.array(dim), feature_type=['cube'],
        feature_coordinates=coords, feature_size=[2],
        signal_magnitude=[1])
    signal = fmrisim.apply_signal(signal_func, vol)

    mask = np.ones(dim)
    noise = fmrisim.generate_noise(
        dimensions=np.array(dim), stimfunction_tr=stim[::int(tr * 100)],
        tr_duration=tr, mask=mask, template=mask * 0.8,
        noise_dict={'sfnr': 60, 'snr': 30})
    brain = 40 * signal + noise

    voxel = brain[5, 5, 5, :]
    r = np.corrcoef(voxel, signal_func[:len(voxel), 0])[0, 1]
    print(f"signal voxel vs HRF-convolved design corr: {r:.2f}")


if __name__ == "__main__":
    main()
