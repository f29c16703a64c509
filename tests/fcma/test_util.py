"""Tests for fcma.util.compute_correlation (ref tests/fcma/test_util.py)."""

import numpy as np

from brainiak_amd.fcma.util import compute_correlation


def test_compute_correlation_matches_corrcoef(seeded_rng):
    a = seeded_rng.randn(8, 20).astype(np.float32)   # [voxels, TRs]
    b = seeded_rng.randn(5, 20).astype(np.float32)
    corr = compute_correlation(a, b)
    ref = np.corrcoef(np.vstack([a, b]))[:8, 8:]
    assert corr.shape == (8, 5)
    assert np.allclose(corr, ref, atol=1e-4)


def test_compute_correlation_zero_variance(seeded_rng):
    a = np.ones((2, 10), dtype=np.float32)
    b = seeded_rng.randn(3, 10).astype(np.float32)
    corr = compute_correlation(a, b)
    assert np.all(np.isfinite(corr))       # zero-var rows -> 0, not NaN
    nan_corr = compute_correlation(a, b, return_nans=True)
    assert np.all(np.isnan(nan_corr[0]))
