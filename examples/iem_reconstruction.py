#!/usr/bin/env python3
"""Inverted encoding model: reconstruct a circular stimulus feature
from synthetic voxel responses (the reference's iem example)."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.reconstruct import InvertedEncoding1D


def main():
    rng = np.random.RandomState(0)
    n_train, n_vox = 120, 50
    angles = rng.rand(n_train) * 180
    # voxels have random circular tuning
    pref = rng.rand(n_vox) * 180
    def resp(a):
        d = np.deg2rad(2 * (a[:, None] - pref[None, :]))
        return np.cos(d) + 0.3 * rng.randn(len(a), n_vox)
    X = resp(angles)

    iem = InvertedEncoding1D(n_channels=6, range_start=0, range_stop=180)
    iem.fit(X, angles)
    test_angles = np.array([20., 65., 110., 155.])
    Xt = resp(test_angles)
    pred = iem.predict(Xt)
    for a, p in zip(test_angles, pred):
        print(f"true {a:6.1f}  reconstructed {p:6.1f}")


if __name__ == "__main__":
    main()
