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


class Classifier:
    """Train/predict on FCMA correlation features with any sklearn-style
    classifier; see module docstring for the contract."""

    def __init__(self, clf, num_processed_voxels=2000, epochs_per_subj=0,
                 device=None):
        self.clf = clf
        self.num_processed_voxels = num_processed_voxels
        self.epochs_per_subj = epochs_per_subj
        self.num_digits_ = 0
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))

    # -- internals ---------------------------------------------------------

    def _stack(self, X_list):
        return torch.stack([
            torch.as_tensor(np.ascontiguousarray(x), dtype=torch.float32)
            for x in X_list]).to(self.device)

    def _prepare_correlation_data(self, X1, X2, start_voxel=0,
                                  num_processed_voxels=None):
        """corr [num_samples, P, V2] for voxels [start, start+P) of X1."""
        num_samples = len(X1)
        assert num_samples > 0, \
            'need at least one (data1, data2) sample to correlate'
        num_voxels1 = X1[0].shape[1]
        num_voxels2 = X2[0].shape[1]
        assert num_voxels1 * num_voxels2 == self.num_features_, \
            'input voxel-pair count differs from the fitted feature count'
        assert X1[0].shape[0] == X2[0].shape[0], \
            'X1 and X2 disagree on TR count'
        if num_processed_voxels is None:
            num_processed_voxels = num_voxels1
        a = self._stack(X1)[:, :, start_voxel:start_voxel
                            + num_processed_voxels]
        b = self._stack(X2)
        return torch.bmm(a.transpose(1, 2), b)

    def _normalize(self, corr, norm_unit):
        """Fisher-z + z-score over each ``norm_unit`` samples (the
        reference reshapes to [1, S, P*V] and calls the N1 kernel)."""
        if norm_unit > 1:
            S, d2, d3 = corr.shape
            flat = corr.reshape(1, S, d2 * d3)
            normalize_correlation_(flat, norm_unit)
            return flat.reshape(S, d2, d3)
        return corr

    @staticmethod
    def _leading_digits(value):
        return len(str(int(value)))

    def _compute_kernel_matrix_in_portion(self, X1, X2):
        kernel = torch.zeros((self.num_samples_, self.num_samples_),
                             dtype=torch.float32, device=self.device)
        sr = 0
        row_length = self.num_processed_voxels
        normalized = None
        while sr < self.num_voxels_:
            if row_length >= self.num_voxels_ - sr:
                row_length = self.num_voxels_ - sr
            corr = self._prepare_correlation_data(X1, X2, sr, row_length)
            normalized = self._normalize(corr, self.epochs_per_subj)
            flat = normalized.reshape(self.num_samples_, -1)
            kernel += flat @ flat.T
            sr += row_length
        self.num_digits_ = self._leading_digits(float(kernel[0, 0]))
        if self.num_digits_ > 2:
            kernel *= 10.0 ** (2 - self.num_digits_)
        return kernel, normalized

    def _generate_training_data(self, X1, X2, num_training_samples):
        if not _is_precomputed_svm(self.clf):
            corr = self._prepare_correlation_data(X1, X2)
            normalized = self._normalize(corr, self.epochs_per_subj)
            data = normalized.reshape(self.num_samples_, self.num_features_)
            self.training_data_ = None
            return data.cpu().numpy()
        if self.num_processed_voxels < self.num_voxels_:
            if num_training_samples is None:
                raise RuntimeError(
                    'the kernel matrix will be computed portion by '
                    'portion, the test samples must be predefined by '
                    'specifying num_training_samples')
            if num_training_samples >= self.num_samples_:
                raise ValueError('the number of training samples must be '
                                 'smaller than the number of total samples')
        data, normalized = self._compute_kernel_matrix_in_portion(X1, X2)
        if self.num_processed_voxels >= self.num_voxels_:
            self.training_data_ = normalized.reshape(
                self.num_samples_, self.num_features_).cpu().numpy()
        else:
            self.training_data_ = None
        return data.cpu().numpy()

    # -- estimator API -----------------------------------------------------

    def fit(self, X, y, num_training_samples=None):
        assert len(X) == len(y), \
            'sample count and label count differ'
        for x in X:
            assert len(x) == 2, \
                'there must be two parts for each correlation computation'
        X1, X2 = zip(*X)
        if not _is_precomputed_svm(self.clf) and \
                num_training_samples is not None:
            num_training_samples = None
            logger.warning(
                'num_training_samples should not be set for classifiers '
                'other than SVM with precomputed kernels')
        num_voxels1 = X1[0].shape[1]
        num_voxels2 = X2[0].shape[1]
        if num_voxels1 < num_voxels2:
            X1, X2 = X2, X1
            num_voxels1, num_voxels2 = num_voxels2, num_voxels1
        self.num_voxels_ = num_voxels1
        self.num_features_ = num_voxels1 * num_voxels2
        self.num_samples_ = len(X1)

        data = self._generate_training_data(X1, X2, num_training_samples)

        if num_training_samples is not None:
            self.test_raw_data_ = None
            self.test_data_ = data[num_training_samples:,
                                   0:num_training_samples]
            data = data[0:num_training_samples, 0:num_training_samples]
        self.clf = self.clf.fit(data, y[0:num_training_samples])
        if num_training_samples is None:
            self.test_raw_data_ = None
            self.test_data_ = None
        return self

    def _build_test_data(self, X):
        for x in X:
            assert len(x) == 2, \
                'there must be two parts for each correlation computation'
        X1, X2 = zip(*X)
        num_voxels1 = X1[0].shape[1]
        num_voxels2 = X2[0].shape[1]
        if num_voxels1 < num_voxels2:
            X1, X2 = X2, X1
            num_voxels1, num_voxels2 = num_voxels2, num_voxels1
        assert self.num_features_ == num_voxels1 * num_voxels2, \
            'the number of features does not match the model'
        num_test_samples = len(X1)
        self.test_raw_data_ = X
        corr = self._prepare_correlation_data(X1, X2)
        normalized = self._normalize(corr, num_test_samples)
        self.test_data_ = self._prepare_test_data(normalized)

    def _prepare_test_data(self, corr_data):
        num_test_samples = corr_data.shape[0]
        assert num_test_samples > 0, 'at least one test sample is needed'
        if _is_precomputed_svm(self.clf):
            assert self.training_data_ is not None, \
                'when using precomputed kernel of SVM, ' \
                'all training data must be provided'
            train = torch.as_tensor(self.training_data_,
                                    device=self.device)
            flat = corr_data.reshape(num_test_samples, self.num_features_)
            data = flat @ train.T
            if self.num_digits_ > 2:
                data = data * 10.0 ** (2 - self.num_digits_)
            return data.cpu().numpy()
        return corr_data.reshape(num_test_samples,
                                 self.num_features_).cpu().numpy()

    def predict(self, X=None):
        if X is not None:
            self._build_test_data(X)
        return self.clf.predict(self.test_data_)

    def _is_equal_to_test_raw_data(self, X):
        if self.test_raw_data_ is None or \
                len(X) != len(self.test_raw_data_):
            return False
        X1, X2 = zip(*X)
        c1, c2 = zip(*self.test_raw_data_)
        for new, old in zip(X1, c1):
            if not np.array_equal(new, old):
                return False
        for new, old in zip(X2, c2):
            if not np.array_equal(new, old):
                return False
        return True

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
