"""Kronecker-product triangular solvers (torch).

Equivalents of the reference's TF kron solvers
(ref src/brainiak/utils/kronecker_solvers.py:6-330; the reference's
``tf_*`` prefix is dropped because these are torch ops — name map:
``tf_kron_mult`` → ``kron_mult``, ``tf_solve_lower_triangular_kron`` →
``solve_lower_triangular_kron``, etc.): solve
``(L_1 ⊗ L_2 ⊗ ... ⊗ L_n) x = y`` for triangular factors without ever
materializing the Kronecker product, by applying per-factor triangular
solves along reshaped axes.  The masked variants handle covariances
restricted to a subset of the Kronecker index set.
"""

from functools import reduce
from typing import List

import torch

__all__ = [
    "kron_mult",
    "masked_triangular_solve",
    "solve_lower_triangular_kron",
    "solve_lower_triangular_masked_kron",
    "solve_upper_triangular_kron",
    "solve_upper_triangular_masked_kron",
]


def _kron_matmul_solve(L: List[torch.Tensor], X: torch.Tensor,
                       upper: bool) -> torch.Tensor:
    """Solve (⊗_i T_i) Z = X with each T_i triangular.

    Uses the identity: solving against a Kronecker product is n
    successive solves, each applied along one tensor axis of X reshaped
    to [n_1, ..., n_k, ncols].
    """
    sizes = [int(m.shape[0]) for m in L]
    n = reduce(lambda a, b: a * b, sizes)
    ncols = X.shape[1]
    Z = X.reshape(sizes + [ncols])
    k = len(L)
    for i, T in enumerate(L):
        # move axis i to the front, flatten the rest
        Zm = torch.movedim(Z, i, 0).reshape(sizes[i], -1)
        Zm = torch.linalg.solve_triangular(T, Zm, upper=upper)
        Zm = Zm.reshape([sizes[i]] + [sizes[j] for j in range(k)
                                      if j != i] + [ncols])
        Z = torch.movedim(Zm, 0, i)
    return Z.reshape(n, ncols)


def kron_mult(L: List[torch.Tensor], X: torch.Tensor) -> torch.Tensor:
    """Compute (⊗_i L_i) @ X without materializing the Kronecker
    product (ref ``tf_kron_mult``)."""
    sizes = [int(m.shape[0]) for m in L]
    n = reduce(lambda a, b: a * b, sizes)
    ncols = X.shape[1]
    Z = X.reshape(sizes + [ncols])
    k = len(L)
    for i, T in enumerate(L):
        Zm = torch.movedim(Z, i, 0).reshape(sizes[i], -1)
        Zm = T @ Zm
        Zm = Zm.reshape([int(T.shape[0])] + [sizes[j] for j in range(k)
                                             if j != i] + [ncols])
        Z = torch.movedim(Zm, 0, i)
    return Z.reshape(n, ncols)


def solve_lower_triangular_kron(L: List[torch.Tensor],
                                X: torch.Tensor) -> torch.Tensor:
    """Solve (⊗ L_i) z = X with lower-triangular factors."""
    return _kron_matmul_solve(L, X, upper=False)


def solve_upper_triangular_kron(L: List[torch.Tensor],
                                X: torch.Tensor) -> torch.Tensor:
    """Solve (⊗ L_i)^T z = X (upper-triangular transposes)."""
    return _kron_matmul_solve([m.T for m in L], X, upper=True)


def masked_triangular_solve(L: torch.Tensor, y: torch.Tensor,
                            mask: torch.Tensor, lower: bool = True,
                            adjoint: bool = False) -> torch.Tensor:
    """Solve L x = y over the mask's valid rows/columns only
    (ref ``tf_masked_triangular_solve``, kronecker_solvers.py:150):
    the solve runs on the mask-selected submatrix and x is scattered
    back with zeros at masked-out rows."""
    idx = torch.nonzero(mask.reshape(-1), as_tuple=True)[0]
    sub = L[idx][:, idx]
    if adjoint:
        sub = sub.T
        lower = not lower
    squeeze = y.dim() == 1
    y2 = y.reshape(-1, 1) if squeeze else y
    xs = torch.linalg.solve_triangular(sub, y2[idx], upper=not lower)
    x = torch.zeros_like(y2)
    x[idx] = xs
    return x.reshape(y.shape)


def _masked_dense(L: List[torch.Tensor], mask: torch.Tensor):
    """Dense masked Kronecker factor: rows/cols of ⊗L_i kept by mask.

    Masked Kronecker structure does not factor into per-axis solves, so
    the masked path materializes the (mask-restricted) matrix — correct
    for the moderate masked sizes these covariances are used at.
    """
    full = L[0]
    for m in L[1:]:
        full = torch.kron(full, m)
    idx = torch.nonzero(mask.reshape(-1), as_tuple=True)[0]
    return full[idx][:, idx]


def solve_lower_triangular_masked_kron(L: List[torch.Tensor],
                                       X: torch.Tensor,
                                       mask: torch.Tensor) -> torch.Tensor:
    dense = _masked_dense(L, mask)
    return torch.linalg.solve_triangular(torch.tril(dense), X,
                                         upper=False)


def solve_upper_triangular_masked_kron(L: List[torch.Tensor],
                                       X: torch.Tensor,
                                       mask: torch.Tensor) -> torch.Tensor:
    dense = _masked_dense(L, mask)
    return torch.linalg.solve_triangular(torch.tril(dense).T, X,
                                         upper=True)
