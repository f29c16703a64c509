"""Kronecker/masked triangular solver checks."""

import torch

from brainiak_amd.utils.kronecker_solvers import masked_triangular_solve

def test_masked_triangular_solve_matches_dense_subsolve():
    torch.manual_seed(4)
    n = 8
    L = torch.tril(torch.rand(n, n) + 2 * torch.eye(n))
    y = torch.rand(n, 3)
    mask = torch.tensor([1, 1, 0, 1, 0, 1, 1, 0])
    idx = mask.nonzero().ravel()
    x = masked_triangular_solve(L, y, mask)
    ref = torch.linalg.solve_triangular(L[idx][:, idx], y[idx],
                                        upper=False)
    assert torch.allclose(x[idx], ref)
    assert torch.all(x[mask == 0] == 0)
    # adjoint solves L^T x = y on the same submatrix
    xa = masked_triangular_solve(L, y, mask, adjoint=True)
    refa = torch.linalg.solve_triangular(L[idx][:, idx].T, y[idx],
                                         upper=True)
    assert torch.allclose(xa[idx], refa)
    # 1-D y keeps its shape
    x1 = masked_triangular_solve(L, y[:, 0], mask)
    assert x1.shape == (n,)
