"""Matrix-normal helpers (torch re-expression of ref
src/brainiak/matnormal/utils.py:86-124 — the reference's TF/tfp
machinery maps onto torch autograd)."""

import numpy as np
import torch
from numpy.linalg import cholesky
from scipy.stats import norm

__all__ = [
    "flatten_cholesky_unique",
    "make_val_and_grad",
    "pack_trainable_vars",
    "rmn",
    "scaled_I",
    "unflatten_cholesky_unique",
    "unpack_trainable_vars",
    "x_tx",
    "xx_t",
]


def rmn(rowcov, colcov):
    """Zero-mean matrix-normal draw with the given row/col covariances."""
    Z = norm.rvs(size=(rowcov.shape[0], colcov.shape[0]))
    return cholesky(rowcov).dot(Z).dot(cholesky(colcov))


def xx_t(x):
    return x @ x.T


def x_tx(x):
    return x.T @ x


def scaled_I(x, size, dtype=torch.float64, device=None):
    return torch.eye(size, dtype=dtype, device=device) * x


def _tril_indices(n, device=None):
    return torch.tril_indices(n, n, device=device)


def flatten_cholesky_unique(L):
    """Pack a (lower-triangular) Cholesky factor into a vector, taking
    log of the diagonal so the parameterization is unique."""
    L = torch.as_tensor(L, dtype=torch.float64)
    n = L.shape[0]
    L = L.clone()
    idx = torch.arange(n)
    L[idx, idx] = torch.log(L[idx, idx])
    rows, cols = _tril_indices(n)
    return L[rows, cols]


def unflatten_cholesky_unique(L_flat):
    """Inverse of flatten_cholesky_unique (diagonal exponentiated)."""
    m = L_flat.shape[0]
    n = int((np.sqrt(8 * m + 1) - 1) / 2)
    L = torch.zeros((n, n), dtype=L_flat.dtype, device=L_flat.device)
    rows, cols = _tril_indices(n, device=L_flat.device)
    L[rows, cols] = L_flat
    idx = torch.arange(n, device=L_flat.device)
    diag = torch.exp(L[idx, idx])
    L = L - torch.diag(torch.diagonal(L)) + torch.diag(diag)
    return L


def pack_trainable_vars(trainable_vars):
    return torch.cat([tv.reshape(-1).detach().cpu()
                      for tv in trainable_vars])


def unpack_trainable_vars(x, trainable_vars):
    sizes = [tuple(tv.shape) for tv in trainable_vars]
    counts = [int(np.prod(sz)) if sz else 1 for sz in sizes]
    flat = torch.split(torch.as_tensor(x, dtype=torch.float64), counts)
    return [fv.reshape(sz) for fv, sz in zip(flat, sizes)]


def make_val_and_grad(lossfn, train_vars):
    """scipy.optimize-compatible (value, gradient) closure via torch
    autograd (the reference builds the same with tf.GradientTape)."""

    def val_and_grad(theta):
        unpacked = unpack_trainable_vars(theta, train_vars)
        for var, val in zip(train_vars, unpacked):
            with torch.no_grad():
                var.copy_(val)
            if var.grad is not None:
                var.grad = None
        loss = lossfn(theta)
        loss.backward()
        grads = []
        for var in train_vars:
            g = var.grad
            grads.append(torch.zeros_like(var) if g is None else g)
        packed = torch.cat([g.reshape(-1) for g in grads])
        return (float(loss.detach().cpu()),
                packed.detach().cpu().numpy())

    return val_and_grad
