"""Correlation-based classification (FCMA stage 2).

API parity with the reference ``Classifier``
(ref src/brainiak/fcma/classifier.py:37-690): same constructor, the same
``fit(X, y, num_training_samples)`` / ``predict`` / ``decision_function``
/ ``score`` contract (X = list of (data1, data2) tuples), the same
portion-by-portion kernel accumulation capped by ``num_processed_voxels``
and the same magnitude-shrink bookkeeping (``num_digits_``).

The BLAS shims the reference drives per sample
(cython_blas.pyx N6/N3/N7) become batched torch matmuls on the selected
device — on MI355X one bmm per portion instead of a Python loop of
sgemm calls, with the Fisher-z normalization running through the same
fused kernel as voxel selection.
"""

import logging

import numpy as np
import torch

from ..utils.timing import stage_timer
from .core import normalize_correlation_

logger = logging.getLogger(__name__)

__all__ = ["Classifier"]


def _is_precomputed_svm(clf):
    try:
        import sklearn.svm
        return isinstance(clf, sklearn.svm.SVC) and \
            clf.kernel == 'precomputed'
    except ImportError:  # pragma: no cover
        return False


def _split_pairs(X):
    """Unzip [(data1, data2), ...] with the WIDER member first (the
    reference orients so voxels1 >= voxels2); returns (X1, X2, v1, v2)."""
    for sample in X:
        if len(sample) != 2:
            raise AssertionError('there must be two parts for each '
                                 'correlation computation')
    left, right = zip(*X)
    v_left, v_right = left[0].shape[1], right[0].shape[1]
    if v_left < v_right:
        return right, left, v_right, v_left
    return left, right, v_left, v_right


class Classifier:
    """Train/predict on FCMA correlation features with any sklearn-style
    classifier; see module docstring for the contract."""

    def __init__(self, clf, num_processed_voxels=2000, epochs_per_subj=0,
                 device=None):
        self.clf = clf
        self.num_processed_voxels = num_processed_voxels
        self.epochs_per_subj = epochs_per_subj
        self.num_digits_ = 0
        if device is not None:
            self.device = torch.device(device)
        else:
            self.device = torch.device(
                "cuda" if torch.cuda.is_available() else "cpu")

    # -- internals ---------------------------------------------------------

    def _stack(self, X_list):
        return torch.stack([
            torch.as_tensor(np.ascontiguousarray(x), dtype=torch.float32)
            for x in X_list]).to(self.device)

    def _prepare_correlation_data(self, X1, X2, start_voxel=0,
                                  num_processed_voxels=None):
        """corr [num_samples, P, V2] for voxels [start, start+P) of X1."""
        if not len(X1):
            raise AssertionError('need at least one (data1, data2) '
                                 'sample to correlate')
        v1, v2 = X1[0].shape[1], X2[0].shape[1]
        if v1 * v2 != self.num_features_:
            raise AssertionError('input voxel-pair count differs from '
                                 'the fitted feature count')
        if X1[0].shape[0] != X2[0].shape[0]:
            raise AssertionError('X1 and X2 disagree on TR count')
        stop = v1 if num_processed_voxels is None \
            else start_voxel + num_processed_voxels
        a = self._stack(X1)[:, :, start_voxel:stop]
        return torch.bmm(a.transpose(1, 2), self._stack(X2))

    def _normalize(self, corr, norm_unit):
        """Fisher-z + z-score over each ``norm_unit`` samples (the
        reference reshapes to [1, S, P*V] and calls the N1 kernel)."""
        if norm_unit <= 1:
            return corr
        S = corr.shape[0]
        flat = corr.reshape(1, S, -1)
        normalize_correlation_(flat, norm_unit)
        return flat.reshape(corr.shape)

    def _apply_shrink(self, kernel):
        """Record the lead-entry digit count and rescale matrices whose
        magnitudes would swamp the SVM (reference num_digits_ logic)."""
        self.num_digits_ = len(str(int(float(kernel[0, 0]))))
        if self.num_digits_ > 2:
            kernel *= 10.0 ** (2 - self.num_digits_)
        return kernel

    def _portioned_kernel(self, X1, X2):
        """Gram matrix accumulated ``num_processed_voxels`` rows at a
        time; returns (kernel, last portion's normalized features)."""
        n = self.num_samples_
        kernel = torch.zeros((n, n), dtype=torch.float32,
                             device=self.device)
        normalized = None
        portions = list(range(0, self.num_voxels_,
                              self.num_processed_voxels))
        for begin in portions:
            width = min(self.num_processed_voxels,
                        self.num_voxels_ - begin)
            corr = self._prepare_correlation_data(X1, X2, begin, width)
            normalized = self._normalize(corr, self.epochs_per_subj)
            flat = normalized.reshape(n, -1)
            kernel += flat @ flat.T
        return self._apply_shrink(kernel), normalized

    def _generate_training_data(self, X1, X2, num_training_samples):
        if not _is_precomputed_svm(self.clf):
            corr = self._prepare_correlation_data(X1, X2)
            features = self._normalize(corr, self.epochs_per_subj)
            self.training_data_ = None
            return features.reshape(self.num_samples_,
                                    self.num_features_).cpu().numpy()
        partial = self.num_processed_voxels < self.num_voxels_
        if partial and num_training_samples is None:
            raise RuntimeError(
                'the kernel matrix will be computed portion by '
                'portion, the test samples must be predefined by '
                'specifying num_training_samples')
        if partial and num_training_samples >= self.num_samples_:
            raise ValueError('the number of training samples must be '
                             'smaller than the number of total samples')
        kernel, normalized = self._portioned_kernel(X1, X2)
        self.training_data_ = None if partial else normalized.reshape(
            self.num_samples_, self.num_features_).cpu().numpy()
        return kernel.cpu().numpy()

    # -- estimator API -----------------------------------------------------

    def fit(self, X, y, num_training_samples=None):
        if len(X) != len(y):
            raise AssertionError('sample count and label count differ')
        X1, X2, v1, v2 = _split_pairs(X)
        if num_training_samples is not None and \
                not _is_precomputed_svm(self.clf):
            num_training_samples = None
            logger.warning(
                'num_training_samples should not be set for classifiers '
                'other than SVM with precomputed kernels')
        self.num_voxels_ = v1
        self.num_features_ = v1 * v2
        self.num_samples_ = len(X1)

        with stage_timer("classifier training data", logger,
                         sync_device=self.device):
            data = self._generate_training_data(X1, X2,
                                                num_training_samples)

        if num_training_samples is not None:
            split = num_training_samples
            self.test_raw_data_ = None
            self.test_data_ = data[split:, :split]
            data = data[:split, :split]
        self.clf = self.clf.fit(data, y[0:num_training_samples])
        if num_training_samples is None:
            self.test_raw_data_ = None
            self.test_data_ = None
        return self

    def _build_test_data(self, X):
        X1, X2, v1, v2 = _split_pairs(X)
        if self.num_features_ != v1 * v2:
            raise AssertionError(
                'the number of features does not match the model')
        self.test_raw_data_ = X
        with stage_timer("classifier test data", logger,
                         sync_device=self.device):
            corr = self._prepare_correlation_data(X1, X2)
            normalized = self._normalize(corr, len(X1))
            self.test_data_ = self._prepare_test_data(normalized)

    def _prepare_test_data(self, corr_data):
        n_test = corr_data.shape[0]
        if n_test < 1:
            raise AssertionError('at least one test sample is needed')
        flat = corr_data.reshape(n_test, self.num_features_)
        if not _is_precomputed_svm(self.clf):
            return flat.cpu().numpy()
        if self.training_data_ is None:
            raise AssertionError('when using precomputed kernel of SVM, '
                                 'all training data must be provided')
        train = torch.as_tensor(self.training_data_, device=self.device)
        similarity = flat @ train.T
        if self.num_digits_ > 2:
            similarity = similarity * 10.0 ** (2 - self.num_digits_)
        return similarity.cpu().numpy()

    def predict(self, X=None):
        if X is not None:
            self._build_test_data(X)
        return self.clf.predict(self.test_data_)

    def _is_equal_to_test_raw_data(self, X):
        cached = self.test_raw_data_
        if cached is None or len(X) != len(cached):
            return False
        return all(np.array_equal(a, b) and np.array_equal(c, d)
                   for (a, c), (b, d) in zip(X, cached))

    def decision_function(self, X=None):
        if X is not None and not self._is_equal_to_test_raw_data(X):
            self._build_test_data(X)
        return self.clf.decision_function(self.test_data_)

    def score(self, X, y, sample_weight=None):
        from sklearn.metrics import accuracy_score
        if _is_precomputed_svm(self.clf) and self.training_data_ is None:
            return accuracy_score(y, self.predict(),
                                  sample_weight=sample_weight)
        return accuracy_score(y, self.predict(X),
                              sample_weight=sample_weight)

    # sklearn BaseEstimator-style params (enables clone/grid-search)
    def get_params(self, deep=True):
        return {"clf": self.clf,
                "num_processed_voxels": self.num_processed_voxels,
                "epochs_per_subj": self.epochs_per_subj}

    def set_params(self, **params):
        for k, v in params.items():
            setattr(self, k, v)
        return self
