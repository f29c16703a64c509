"""Correlation helpers (API parity: ref src/brainiak/fcma/util.py).

``compute_correlation`` reduces Pearson correlation to one fp32 GEMM;
on GPU tensors the matmul runs on the MFMA path through rocBLAS.
"""

import numpy as np
import torch

__all__ = ["compute_correlation"]


def _unitize_rows(m, return_nans=False):
    """Center each row, scale to unit norm x 1/sqrt(n): after this,
    correlation is a plain dot product.  Zero-variance rows become 0
    (or NaN when return_nans=True)."""
    centered = m - m.mean(axis=1, keepdims=True)
    sd = centered.std(axis=1, keepdims=True)
    with np.errstate(invalid="ignore", divide="ignore"):
        unit = centered / (sd * np.sqrt(m.shape[1]))
    return unit if return_nans else np.nan_to_num(unit)


def compute_correlation(matrix1, matrix2, return_nans=False):
    """Pearson correlation of the rows of ``matrix1`` [r1, c] with the
    rows of ``matrix2`` [r2, c] → fp32 [r1, r2].

    Vectors with zero variance yield 0 (or NaN when return_nans=True).
    """
    matrix1 = np.asarray(matrix1, dtype=np.float32)
    matrix2 = np.asarray(matrix2, dtype=np.float32)
    if matrix1.shape[1] != matrix2.shape[1]:
        raise ValueError('Dimension discrepancy')
    t1 = torch.from_numpy(
        np.ascontiguousarray(_unitize_rows(matrix1, return_nans)))
    t2 = torch.from_numpy(
        np.ascontiguousarray(_unitize_rows(matrix2, return_nans)))
    return (t1 @ t2.T).numpy().astype(np.float32, copy=False)
