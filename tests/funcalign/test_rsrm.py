import numpy as np
import pytest

from brainiak_amd.funcalign.rsrm import RSRM
from brainiak_amd.funcalign.srm import NotFittedError


def _rsrm_data(rng, subjects=3, voxels=40, trs=30, features=4):
    R = rng.randn(features, trs)
    X, W = [], []
    for _ in range(subjects):
        q, _ = np.linalg.qr(rng.randn(voxels, features))
        # sparse individual outliers
        S = np.zeros((voxels, trs))
        idx = rng.choice(voxels * trs, 20, replace=False)
        S.ravel()[idx] = rng.randn(20) * 5
        X.append(q @ R + S + 0.05 * rng.randn(voxels, trs))
        W.append(q)
    return X, W, R


def test_rsrm_fit(seeded_rng):
    X, _, R_true = _rsrm_data(seeded_rng)
    model = RSRM(n_iter=15, features=4, gamma=1.0, rand_seed=0,
                 device="cpu")
    model.fit(X)
    assert len(model.w_) == 3
    for w in model.w_:
        assert np.allclose(w.T @ w, np.eye(4), atol=1e-8)
    assert model.r_.shape == (4, 30)
    # individual terms should be sparse
    sparsity = np.mean([np.mean(s == 0) for s in model.s_])
    assert sparsity > 0.5
    # shared responses aligned across subjects
    r, s = model.transform(X)
    c = np.corrcoef(r[0].ravel(), r[1].ravel())[0, 1]
    assert c > 0.9


def test_rsrm_validation(seeded_rng):
    X, _, _ = _rsrm_data(seeded_rng)
    with pytest.raises(ValueError):
        RSRM(gamma=-1.0, device="cpu").fit(X)
    with pytest.raises(ValueError):
        RSRM(features=4, device="cpu").fit([X[0]])
    with pytest.raises(ValueError):
        RSRM(features=50, device="cpu").fit(X)
    with pytest.raises(NotFittedError):
        RSRM(device="cpu").transform(X)


def test_rsrm_transform_subject(seeded_rng):
    X, _, _ = _rsrm_data(seeded_rng, subjects=4)
    model = RSRM(n_iter=12, features=4, gamma=1.0, device="cpu")
    model.fit(X[:3])
    w, s = model.transform_subject(X[3])
    assert w.shape == (40, 4)
    assert np.allclose(w.T @ w, np.eye(4), atol=1e-6)
    with pytest.raises(ValueError):
        model.transform_subject(X[3][:, :10])


def test_shrink():
    import torch
    v = torch.tensor([-3.0, -0.5, 0.0, 0.5, 3.0])
    out = RSRM._shrink(v.clone(), 1.0)
    assert torch.allclose(out, torch.tensor([-2.0, 0.0, 0.0, 0.0, 2.0]))
