"""Correlation helpers (API parity: ref src/brainiak/fcma/util.py).

``compute_correlation`` reduces Pearson correlation to one fp32 GEMM;
on GPU tensors the matmul runs on the MFMA path through rocBLAS.
"""

import math

import numpy as np
import torch
from scipy.stats import zscore

__all__ = ["compute_correlation"]


def _normalize_for_correlation(data, axis, return_nans=False):
    """Z-score (ddof=0) along ``axis`` then scale by 1/sqrt(n) so that
    correlation becomes a plain dot product."""
    shape = data.shape
    data = zscore(data, axis=axis, ddof=0)
    if not return_nans:
        data = np.nan_to_num(data)
    return data / math.sqrt(shape[axis])


def compute_correlation(matrix1, matrix2, return_nans=False):
    """Pearson correlation of the rows of ``matrix1`` [r1, c] with the
    rows of ``matrix2`` [r2, c] → fp32 [r1, r2].

    Vectors with zero variance yield 0 (or NaN when return_nans=True).
    """
    matrix1 = np.asarray(matrix1, dtype=np.float32)
    matrix2 = np.asarray(matrix2, dtype=np.float32)
    r1, d1 = matrix1.shape
    r2, d2 = matrix2.shape
    if d1 != d2:
        raise ValueError('Dimension discrepancy')
    m1 = _normalize_for_correlation(matrix1, 1, return_nans=return_nans)
    m2 = _normalize_for_correlation(matrix2, 1, return_nans=return_nans)
    t1 = torch.from_numpy(np.ascontiguousarray(m1))
    t2 = torch.from_numpy(np.ascontiguousarray(m2))
    return (t1 @ t2.T).numpy().astype(np.float32, copy=False)
