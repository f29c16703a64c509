"""Robust Shared Response Model (RSRM).

API parity with the reference (ref src/brainiak/funcalign/rsrm.py:39-561):
X_i ≈ W_i R + S_i with orthogonal W_i, shared response R, and sparse
individual terms S_i (L1, soft-threshold shrinkage), fit by block
coordinate descent.

Math runs on torch (rocBLAS gemms + the Gram-based Procrustes polar
factor shared with SRM) on the configured device.

Citation: [Turek2017] "A semi-supervised method for multi-subject fMRI
functional alignment", ICASSP 2017 (robust variant).
"""

import logging

import numpy as np
import torch

from .srm import NotFittedError, _polar_orthogonal, _to_tensor

logger = logging.getLogger(__name__)

__all__ = ["RSRM"]


class RSRM:
    """Robust SRM via BCD; parameters n_iter, features, gamma, rand_seed
    (+ device).  Attributes after fit: w_ (list [V_i, K]), r_ [K, T],
    s_ (list [V_i, T]), random_state_."""

    def __init__(self, n_iter=10, features=50, gamma=1.0, rand_seed=0,
                 device=None):
        self.n_iter = n_iter
        self.features = features
        self.gamma = gamma
        self.rand_seed = rand_seed
        self.device = device

    def _dev_dtype(self):
        dev = torch.device(self.device) if self.device else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        return dev, (torch.float64 if dev.type == "cpu" else torch.float32)

    def fit(self, X, y=None):
        logger.info('Starting RSRM')
        if 0.0 >= self.gamma:
            raise ValueError("Gamma parameter should be positive.")
        if len(X) <= 1:
            raise ValueError("There are not enough subjects in the input "
                             "data to train the model.")
        if X[0].shape[1] < self.features:
            raise ValueError(
                "There are not enough timepoints to train the model with "
                "{0:d} features.".format(self.features))
        number_trs = X[0].shape[1]
        for subject in range(len(X)):
            if not np.all(np.isfinite(np.asarray(X[subject]))):
                raise ValueError("Input contains NaN or infinity.")
            if X[subject].shape[1] != number_trs:
                raise ValueError("Different number of alignment timepoints "
                                 "between subjects.")
        self.random_state_ = np.random.RandomState(self.rand_seed)
        self.w_, self.r_, self.s_ = self._rsrm(X)
        return self

    def transform(self, X):
        """Returns (r, s): per-subject shared responses and individual
        (sparse) terms for new data."""
        if not hasattr(self, 'w_'):
            raise NotFittedError("The model fit has not been run yet.")
        if len(X) != len(self.w_):
            raise ValueError("The number of subjects does not match the one"
                             " in the model.")
        r = [None] * len(X)
        s = [None] * len(X)
        for subject in range(len(X)):
            if X[subject] is not None:
                r[subject], s[subject] = self._transform_new_data(
                    X[subject], subject)
        return r, s

    def _transform_new_data(self, X, subject):
        dev, dtype = self._dev_dtype()
        x = _to_tensor(X, dev, dtype)
        w = _to_tensor(self.w_[subject], dev, dtype)
        s = torch.zeros_like(x)
        r = None
        for _ in range(self.n_iter):
            r = w.T @ (x - s)
            s = self._shrink(x - w @ r, self.gamma)
        return r.cpu().numpy(), s.cpu().numpy()

    def transform_subject(self, X):
        """Returns (w, s) for a new subject given the fitted shared R."""
        if not hasattr(self, 'w_'):
            raise NotFittedError("The model fit has not been run yet.")
        if X.shape[1] != self.r_.shape[1]:
            raise ValueError("The number of timepoints(TRs) does not match "
                             "the one in the model.")
        dev, dtype = self._dev_dtype()
        x = _to_tensor(X, dev, dtype)
        r = _to_tensor(self.r_, dev, dtype)
        s = torch.zeros_like(x)
        w = None
        for _ in range(self.n_iter):
            w = _polar_orthogonal((x - s) @ r.T, perturb=0.0)
            s = self._shrink(x - w @ r, self.gamma)
        return w.cpu().numpy(), s.cpu().numpy()

    def _rsrm(self, X):
        dev, dtype = self._dev_dtype()
        subjs = len(X)
        x = [_to_tensor(d, dev, dtype) for d in X]
        voxels = [d.shape[0] for d in x]
        TRs = x[0].shape[1]

        W = []
        for i in range(subjs):
            rnd = self.random_state_.random_sample((voxels[i],
                                                    self.features))
            q, _ = np.linalg.qr(rnd)
            W.append(_to_tensor(q, dev, dtype))
        S = [torch.zeros((voxels[i], TRs), dtype=dtype, device=dev)
             for i in range(subjs)]
        R = self._update_shared_response(x, S, W, self.features)

        if logger.isEnabledFor(logging.INFO):
            logger.info('Objective function %f',
                        self._objective_function(x, W, R, S, self.gamma))
        for _ in range(self.n_iter):
            W = [_polar_orthogonal((x[i] - S[i]) @ R.T, perturb=0.0)
                 for i in range(subjs)]
            S = [self._shrink(x[i] - W[i] @ R, self.gamma)
                 for i in range(subjs)]
            R = self._update_shared_response(x, S, W, self.features)
            if logger.isEnabledFor(logging.INFO):
                logger.info('Objective function %f',
                            self._objective_function(x, W, R, S,
                                                     self.gamma))
        return ([w.cpu().numpy() for w in W], R.cpu().numpy(),
                [s.cpu().numpy() for s in S])

    @staticmethod
    def _objective_function(X, W, R, S, gamma):
        func = 0.0
        for i in range(len(X)):
            func += 0.5 * float(((X[i] - W[i] @ R - S[i]) ** 2).sum()) \
                + gamma * float(S[i].abs().sum())
        return func

    @staticmethod
    def _update_shared_response(X, S, W, features):
        R = W[0].T @ (X[0] - S[0])
        for i in range(1, len(X)):
            R = R + W[i].T @ (X[i] - S[i])
        return R / len(X)

    @staticmethod
    def _shrink(v, gamma):
        """Soft-threshold shrinkage (elementwise)."""
        return torch.sign(v) * (v.abs() - gamma).clamp_min(0.0)

    def get_params(self, deep=True):
        return {"n_iter": self.n_iter, "features": self.features,
                "gamma": self.gamma, "rand_seed": self.rand_seed}

    def set_params(self, **params):
        for k, v in params.items():
            setattr(self, k, v)
        return self
