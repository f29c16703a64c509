"""Semi-Supervised Shared Response Model (SS-SRM).

API parity with the reference (ref src/brainiak/funcalign/sssrm.py:55-822):
block coordinate descent over per-subject orthogonal maps W_i (Stiefel
manifold), the shared response S, and a multinomial logistic-regression
classifier (theta, bias), optimizing

    (1-α)·Loss_SRM(W, S; X) + α/γ·Loss_MLR(θ, b; WᵀZ, y) + ‖θ‖²/2

The reference builds its cost graphs in TensorFlow and optimizes W with
pymanopt's Stiefel conjugate gradient; here both are torch autograd —
W via Riemannian gradient descent with QR retraction and backtracking
line search, (θ, b) via scipy L-BFGS.

Citation: [Turek2016] "A semi-supervised method for multi-subject fMRI
functional alignment", ICASSP 2017.
"""

import logging

import numpy as np
import torch
from scipy.optimize import minimize

from ..utils.utils import concatenate_not_none
from .srm import NotFittedError, _init_w

logger = logging.getLogger(__name__)

__all__ = ["SSSRM"]

_DT = torch.float64


class SSSRM:
    """SS-SRM estimator; constructor matches the reference
    (n_iter, features, gamma, alpha, rand_seed)."""

    def __init__(self, n_iter=10, features=50, gamma=1.0, alpha=0.5,
                 rand_seed=0):
        self.n_iter = n_iter
        self.features = features
        self.gamma = gamma
        self.alpha = alpha
        self.rand_seed = rand_seed

    # -- API ---------------------------------------------------------------

    def fit(self, X, y, Z):
        """X: alignment data (list of [V_i, n_align]); y: label lists;
        Z: classification data (list of [V_i, samples_i])."""
        logger.info('Starting SS-SRM')
        if 0.0 >= self.alpha or self.alpha >= 1.0:
            raise ValueError(
                "Alpha parameter should be in range (0.0, 1.0)")
        if 0.0 >= self.gamma:
            raise ValueError("Gamma parameter should be positive.")
        if len(X) <= 1 or len(y) <= 1 or len(Z) <= 1:
            raise ValueError("There are not enough subjects in the input "
                             "data to train the model.")
        if len(X) != len(y) or len(X) != len(Z):
            raise ValueError("Different number of subjects in data.")
        if X[0].shape[1] < self.features:
            raise ValueError(
                "There are not enough samples to train the model with "
                "{0:d} features.".format(self.features))
        number_trs = X[0].shape[1]
        for s in range(len(X)):
            if not np.all(np.isfinite(X[s])) or \
                    not np.all(np.isfinite(Z[s])):
                raise ValueError("Input contains NaN or infinity.")
            if X[s].shape[1] != number_trs:
                raise ValueError("Different number of alignment samples "
                                 "between subjects.")
            if X[s].shape[0] != Z[s].shape[0]:
                raise ValueError(
                    "Different number of voxels between alignment and "
                    "classification data (subject {0:d}).".format(s))
            if Z[s].shape[1] != np.size(y[s]):
                raise ValueError("Different number of samples and labels "
                                 "in subject {0:d}.".format(s))

        new_y = self._init_classes(y)
        self.w_, self.s_, self.theta_, self.bias_ = self._sssrm(
            X, Z, new_y)
        return self

    def _init_classes(self, y):
        self.classes_ = np.unique(concatenate_not_none(
            [np.asarray(v) for v in y]))
        return [np.digitize(np.asarray(v), self.classes_) - 1 for v in y]

    def transform(self, X, y=None):
        """Project each subject's data to the shared space."""
        if not hasattr(self, 'w_'):
            raise NotFittedError("The model fit has not been run yet.")
        if len(X) != len(self.w_):
            raise ValueError("The number of subjects does not match the "
                             "one in the model.")
        return [self.w_[i].T.dot(X[i]) if X[i] is not None else None
                for i in range(len(X))]

    def predict(self, X):
        """MLR prediction in the shared space, per subject."""
        if not hasattr(self, 'w_'):
            raise NotFittedError("The model fit has not been run yet.")
        if len(X) != len(self.w_):
            raise ValueError("The number of subjects does not match the "
                             "one in the model.")
        X_shared = self.transform(X)
        p = [None] * len(X_shared)
        for s in range(len(X_shared)):
            if X_shared[s] is not None:
                scores = self.theta_.T.dot(X_shared[s]) + \
                    self.bias_[:, np.newaxis]
                p[s] = self.classes_[scores.argmax(axis=0)]
        return p

    # -- BCD ---------------------------------------------------------------

    def _sssrm(self, data_align, data_sup, labels):
        classes = self.classes_.size
        self.random_state_ = np.random.RandomState(self.rand_seed)
        random_states = [
            np.random.RandomState(
                self.random_state_.randint(2 ** 32, dtype=np.int64))
            for _ in range(len(data_align))]

        from ..parallel import DistContext
        serial = DistContext.__new__(DistContext)
        serial.rank, serial.world_size = 0, 1
        serial.backend, serial._owns_group = None, False
        serial.device = torch.device("cpu")
        w_np, _ = _init_w(data_align, self.features, random_states, serial)
        w = [torch.as_tensor(wi, dtype=_DT) for wi in w_np]
        X = [torch.as_tensor(np.asarray(d), dtype=_DT)
             for d in data_align]
        Z = [torch.as_tensor(np.asarray(d), dtype=_DT) for d in data_sup]
        y = [torch.as_tensor(np.asarray(v), dtype=torch.long)
             for v in labels]

        s = self._shared(X, w)
        theta, bias = self._update_classifier(Z, y, w, classes)

        for iteration in range(self.n_iter):
            logger.info('Iteration %d', iteration + 1)
            w = [self._update_w_subject(X[i], Z[i], y[i], w[i], s, theta,
                                        bias) for i in range(len(X))]
            s = self._shared(X, w)
            theta, bias = self._update_classifier(Z, y, w, classes)
            if logger.isEnabledFor(logging.INFO):
                logger.info('Objective function %f',
                            self._objective(X, Z, y, w, s, theta, bias))

        return ([wi.numpy() for wi in w], s.numpy(), theta.numpy(),
                bias.numpy())

    @staticmethod
    def _shared(X, w):
        s = w[0].T @ X[0]
        for m in range(1, len(w)):
            s = s + w[m].T @ X[m]
        return s / len(w)

    def _mlr_loss_subject(self, Zi, yi, wi, theta, bias):
        """α/(n γ)·cross-entropy of the MLR on subject i's projected
        data (no θ regularization)."""
        if Zi is None or yi.numel() == 0:
            return torch.zeros((), dtype=_DT)
        logits = theta.T @ (wi.T @ Zi) + bias[:, None]   # [classes, n]
        n = Zi.shape[1]
        ce = torch.logsumexp(logits, dim=0).sum() - \
            logits[yi, torch.arange(n)].sum()
        return self.alpha / n / self.gamma * ce

    def _srm_loss_subject(self, Xi, wi, s):
        n = Xi.shape[1]
        return (1 - self.alpha) * 0.5 / n * ((Xi - wi @ s) ** 2).sum()

    def _objective(self, X, Z, y, w, s, theta, bias):
        obj = 0.5 * (theta ** 2).sum()
        for i in range(len(X)):
            obj = obj + self._srm_loss_subject(X[i], w[i], s) + \
                self._mlr_loss_subject(Z[i], y[i], w[i], theta, bias)
        return float(obj)

    def _update_w_subject(self, Xi, Zi, yi, wi, s, theta, bias,
                          steps=30, lr=None):
        """Riemannian gradient descent on the Stiefel manifold with QR
        retraction + backtracking (replaces pymanopt's Stiefel CG)."""
        W = wi.clone()

        def f(Wt):
            return self._srm_loss_subject(Xi, Wt, s) + \
                self._mlr_loss_subject(Zi, yi, Wt, theta, bias)

        fw = f(W)
        if lr is None:
            lr = 1.0
        for _ in range(steps):
            Wp = W.clone().requires_grad_(True)
            loss = f(Wp)
            loss.backward()
            G = Wp.grad
            # Riemannian gradient: project onto the tangent space
            WtG = W.T @ G
            rgrad = G - W @ ((WtG + WtG.T) / 2)
            gnorm = float(rgrad.norm())
            if gnorm < 1e-10:
                break
            # backtracking line search with QR retraction
            improved = False
            t = lr
            for _bt in range(20):
                Q, R = torch.linalg.qr(W - t * rgrad)
                # fix QR sign ambiguity for a proper retraction
                Q = Q * torch.sign(torch.diagonal(R)).clamp(min=-1)[None, :]
                with torch.no_grad():
                    fn = f(Q)
                if float(fn) < float(fw) - 1e-4 * t * gnorm ** 2:
                    W, fw = Q, fn
                    lr = t * 1.5
                    improved = True
                    break
                t *= 0.5
            if not improved:
                break
        return W

    def _update_classifier(self, Z, y, w, classes):
        """Weighted, L2-regularized MLR over all subjects' projected
        samples (scipy L-BFGS over torch autograd)."""
        feats = self.features
        proj, labels, weights = [], [], []
        for i in range(len(Z)):
            if Z[i] is not None and y[i].numel():
                proj.append(w[i].T @ Z[i])
                labels.append(y[i])
                weights.append(torch.full((Z[i].shape[1],),
                                          float(Z[i].shape[1]), dtype=_DT))
        data = torch.cat(proj, dim=1)          # [K, n_total]
        yy = torch.cat(labels)
        ww = torch.cat(weights)

        n_par = classes * feats + classes
        theta0 = np.zeros(n_par)

        def val_and_grad(packed):
            t = torch.as_tensor(packed, dtype=_DT).requires_grad_(True)
            theta = t[:classes * feats].reshape(classes, feats)
            bias = t[classes * feats:]
            logits = theta @ data + bias[:, None]
            ce_per = torch.logsumexp(logits, dim=0) - \
                logits[yy, torch.arange(data.shape[1])]
            loss = (self.alpha / self.gamma) * (ce_per / ww).sum() \
                + 0.5 * (theta ** 2).sum()
            loss.backward()
            return float(loss.detach()), t.grad.numpy()

        res = minimize(val_and_grad, theta0, jac=True, method='L-BFGS-B',
                       options={'maxiter': 200})
        theta = torch.as_tensor(
            res.x[:classes * feats].reshape(classes, feats).T.copy(),
            dtype=_DT)  # stored as [features, classes] like the reference
        bias = torch.as_tensor(res.x[classes * feats:], dtype=_DT)
        return theta, bias

    def get_params(self, deep=True):
        return {"n_iter": self.n_iter, "features": self.features,
                "gamma": self.gamma, "alpha": self.alpha,
                "rand_seed": self.rand_seed}

    def set_params(self, **params):
        for k, v in params.items():
            setattr(self, k, v)
        return self
