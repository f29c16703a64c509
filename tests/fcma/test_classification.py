import numpy as np
import pytest
from sklearn import svm

from brainiak_amd.fcma.classifier import Classifier


def _samples(rng, n_samples=16, trs=12, v1=20, v2=10, signal=True):
    """List of (data1, data2) tuples + labels with condition-dependent
    cross-ROI correlation."""
    X, y = [], []
    for i in range(n_samples):
        cond = i % 2
        d1 = rng.randn(trs, v1).astype(np.float32)
        d2 = rng.randn(trs, v2).astype(np.float32)
        if signal and cond == 0:
            shared = rng.randn(trs, 1)
            d1[:, :6] += 2.5 * shared
            d2[:, :4] += 2.5 * shared
        X.append((d1, d2))
        y.append(cond)
    return X, np.asarray(y)


def test_classifier_precomputed_kernel_roundtrip(seeded_rng):
    X, y = _samples(seeded_rng)
    clf = Classifier(svm.SVC(kernel='precomputed', shrinking=False, C=1.0),
                     epochs_per_subj=4, device="cpu")
    clf.fit(X, y)
    assert clf.num_voxels_ == 20
    assert clf.num_features_ == 200
    pred = clf.predict(X)
    assert pred.shape == (16,)
    assert clf.score(X, y) > 0.8
    dec = clf.decision_function(X)
    assert dec.shape == (16,)
    assert np.array_equal(pred == 1, dec > 0) or \
        np.array_equal(pred == 1, dec < 0)


def test_classifier_portioned_kernel_matches_full(seeded_rng):
    X, y = _samples(seeded_rng)
    full = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                      epochs_per_subj=4, device="cpu")
    full.fit(X, y)

    portioned = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                           num_processed_voxels=7, epochs_per_subj=4,
                           device="cpu")
    portioned.fit(X, y, num_training_samples=12)
    # the portioned path stores test rows of the same kernel: scores on
    # the held-out samples must be identical to re-deriving from full
    pred = portioned.predict()
    assert pred.shape == (4,)
    acc = portioned.score(None, y[12:])
    assert 0.0 <= acc <= 1.0


def test_classifier_portioned_requires_num_training_samples(seeded_rng):
    X, y = _samples(seeded_rng, n_samples=8)
    c = Classifier(svm.SVC(kernel='precomputed'), num_processed_voxels=5,
                   epochs_per_subj=4, device="cpu")
    with pytest.raises(RuntimeError):
        c.fit(X, y)
    with pytest.raises(ValueError):
        c.fit(X, y, num_training_samples=8)


def test_classifier_non_svm(seeded_rng):
    from sklearn.linear_model import LogisticRegression
    X, y = _samples(seeded_rng, n_samples=12)
    c = Classifier(LogisticRegression(max_iter=200), epochs_per_subj=4,
                   device="cpu")
    c.fit(X, y)
    assert c.training_data_ is None
    assert c.predict(X).shape == (12,)
    assert c.score(X, y) >= 0.5


def test_classifier_swaps_larger_mask_first(seeded_rng):
    """When data2 has more voxels the classifier swaps internally."""
    X, y = _samples(seeded_rng, v1=8, v2=15)
    c = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                   epochs_per_subj=4, device="cpu")
    c.fit(X, y)
    assert c.num_voxels_ == 15
    assert c.num_features_ == 120


# -- round-2 depth (ref tests/fcma/test_classification.py:1-222) -------------

def test_classifier_train_test_split_path(seeded_rng):
    """num_training_samples: the kernel rows for the held-out samples
    are cached at fit time and predict() consumes them without
    recomputing (the reference's test_raw_data_ caching contract)."""
    X, y = _samples(seeded_rng, n_samples=20)
    clf = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                     epochs_per_subj=4, device="cpu")
    clf.fit(X, y, num_training_samples=16)
    assert clf.test_data_ is not None
    assert clf.test_data_.shape == (4, 16)
    pred = clf.predict()              # cached path, no X argument
    assert pred.shape == (4,)
    acc = clf.score(None, y[16:])
    assert 0.0 <= acc <= 1.0


def test_classifier_ignores_split_for_non_precomputed(seeded_rng):
    """num_training_samples only applies to precomputed kernels — the
    reference warns and ignores it (classifier.py:146-149)."""
    X, y = _samples(seeded_rng)
    clf = Classifier(svm.SVC(kernel='rbf'), epochs_per_subj=4,
                     device="cpu")
    clf.fit(X, y, num_training_samples=12)
    assert clf.test_data_ is None     # split was dropped


def test_classifier_non_precomputed_kernel(seeded_rng):
    """A non-precomputed sklearn kernel trains on the normalized
    correlation features directly."""
    X, y = _samples(seeded_rng)
    clf = Classifier(svm.SVC(kernel='linear'), epochs_per_subj=4,
                     device="cpu")
    clf.fit(X, y)
    assert clf.score(X, y) > 0.7


def test_classifier_self_correlation_single_mask(seeded_rng):
    """X as a list of single arrays → self-correlation features."""
    rng = seeded_rng
    X, y = [], []
    for i in range(16):
        d = rng.randn(12, 14).astype(np.float32)
        if i % 2 == 0:
            d[:, :5] += 2.0 * rng.randn(12, 1)
        X.append(d)
        y.append(i % 2)
    clf = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                     epochs_per_subj=4, device="cpu")
    clf.fit(list(zip(X, X)), np.asarray(y))
    assert clf.num_features_ == 14 * 14
    assert clf.score(list(zip(X, X)), np.asarray(y)) > 0.7


def test_classifier_mismatched_labels_raise(seeded_rng):
    X, y = _samples(seeded_rng)
    clf = Classifier(svm.SVC(kernel='precomputed'), epochs_per_subj=4,
                     device="cpu")
    with pytest.raises(AssertionError):
        clf.fit(X, y[:-3])


def test_classifier_portioned_kernel_matches_direct(seeded_rng):
    """Portion-accumulated kernel build (num_processed_voxels < V1)
    gives the same decisions as the single-portion build
    (ref classifier.py:279-348)."""
    X, y = _samples(seeded_rng, n_samples=20, v1=30)
    a = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                   epochs_per_subj=4, device="cpu",
                   num_processed_voxels=7)
    b = Classifier(svm.SVC(kernel='precomputed', shrinking=False),
                   epochs_per_subj=4, device="cpu")
    # the portioned path requires a predefined train/test split
    # (ref classifier.py:117-124)
    a.fit(X, y, num_training_samples=16)
    b.fit(X, y, num_training_samples=16)
    # identical held-out kernel rows from both accumulation orders
    assert np.allclose(a.test_data_, b.test_data_, atol=1e-4)
    assert np.array_equal(a.predict(), b.predict())
    # and the portioned path REFUSES to run without the split
    with pytest.raises(RuntimeError):
        Classifier(svm.SVC(kernel='precomputed'), epochs_per_subj=4,
                   device="cpu", num_processed_voxels=7).fit(X, y)
