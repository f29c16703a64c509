import numpy as np
import pytest

from brainiak_amd.funcalign import srm as srm_mod
from brainiak_amd.funcalign.srm import SRM, DetSRM, load


def _synthetic_subjects(rng, subjects=4, voxels=60, samples=40, features=5,
                        noise=0.1):
    """Shared low-rank latent + per-subject orthogonal mixing + noise."""
    S = rng.randn(features, samples)
    data, W = [], []
    for _ in range(subjects):
        q, _ = np.linalg.qr(rng.randn(voxels, features))
        W.append(q)
        data.append(q @ S + noise * rng.randn(voxels, samples))
    return data, W, S


def test_detsrm_fit_recovers_shared_space(seeded_rng):
    data, _, S = _synthetic_subjects(seeded_rng)
    model = DetSRM(n_iter=15, features=5, rand_seed=0, device="cpu")
    model.fit(data)
    assert len(model.w_) == 4
    for w in model.w_:
        assert w.shape == (60, 5)
        # orthogonality
        assert np.allclose(w.T @ w, np.eye(5), atol=1e-8)
    assert model.s_.shape == (5, 40)
    # transformed data should correlate across subjects much more than raw
    projected = model.transform(data)
    c = np.corrcoef(projected[0].ravel(), projected[1].ravel())[0, 1]
    assert abs(c) > 0.9


def test_detsrm_objective_decreases(seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng)
    m2 = DetSRM(n_iter=2, features=5, rand_seed=0, device="cpu").fit(data)
    m10 = DetSRM(n_iter=12, features=5, rand_seed=0, device="cpu").fit(data)
    obj2 = m2._objective_function(data, m2.w_, m2.s_)
    obj10 = m10._objective_function(data, m10.w_, m10.s_)
    assert obj10 <= obj2 + 1e-9


def test_srm_fit_basic(seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng, subjects=3)
    model = SRM(n_iter=8, features=5, rand_seed=0, device="cpu")
    model.fit(data)
    assert model.s_.shape == (5, 40)
    assert model.sigma_s_.shape == (5, 5)
    assert model.rho2_.shape == (3,)
    assert np.all(model.rho2_ > 0)
    for w, mu in zip(model.w_, model.mu_):
        assert w.shape == (60, 5)
        assert np.allclose(w.T @ w, np.eye(5), atol=1e-8)
        assert mu.shape == (60,)
    # eigenvalues of sigma_s positive
    assert np.all(np.linalg.eigvalsh(model.sigma_s_) > -1e-10)


def test_srm_errors(seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng, subjects=2, samples=40)
    with pytest.raises(ValueError):
        SRM(device="cpu").fit([data[0]])  # too few subjects
    with pytest.raises(ValueError):
        SRM(features=50, device="cpu").fit(data)  # samples < features
    bad = [data[0], data[1][:, :30]]
    with pytest.raises(ValueError):
        SRM(features=5, device="cpu").fit(bad)  # unequal samples
    nan_data = [d.copy() for d in data]
    nan_data[0][0, 0] = np.nan
    with pytest.raises(ValueError):
        SRM(features=5, device="cpu").fit(nan_data)
    with pytest.raises(srm_mod.NotFittedError):
        SRM(device="cpu").transform(data)


def test_srm_transform_and_new_subject(seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng, subjects=4)
    model = SRM(n_iter=8, features=5, rand_seed=0, device="cpu").fit(data[:3])
    with pytest.raises(ValueError):
        model.transform(data)  # 4 subjects vs model's 3
    proj = model.transform(data[:3])
    assert proj[0].shape == (5, 40)
    w_new = model.transform_subject(data[3])
    assert w_new.shape == (60, 5)
    assert np.allclose(w_new.T @ w_new, np.eye(5), atol=1e-6)
    with pytest.raises(ValueError):
        model.transform_subject(data[3][:, :20])


def test_srm_save_load_roundtrip(tmp_path, seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng, subjects=3)
    model = SRM(n_iter=5, features=5, rand_seed=3, device="cpu").fit(data)
    f = tmp_path / "model.npz"
    model.save(f)
    loaded = load(f)
    assert loaded.features == 5 and loaded.n_iter == 5
    assert np.allclose(loaded.s_, model.s_)
    assert np.allclose(loaded.sigma_s_, model.sigma_s_)
    for a, b in zip(loaded.w_, model.w_):
        assert np.allclose(a, b)


def test_srm_seed_determinism(seeded_rng):
    data, _, _ = _synthetic_subjects(seeded_rng, subjects=3)
    m1 = SRM(n_iter=4, features=5, rand_seed=7, device="cpu").fit(data)
    m2 = SRM(n_iter=4, features=5, rand_seed=7, device="cpu").fit(data)
    assert np.allclose(m1.s_, m2.s_)
    m3 = SRM(n_iter=4, features=5, rand_seed=8, device="cpu").fit(data)
    assert not np.allclose(m1.w_[0], m3.w_[0])


def test_polar_orthogonal_matches_svd(seeded_rng):
    import torch
    A = torch.tensor(seeded_rng.randn(30, 5))
    W = srm_mod._polar_orthogonal(A, perturb=0.0).numpy()
    U, _, Vt = np.linalg.svd(A.numpy(), full_matrices=False)
    assert np.allclose(W, U @ Vt, atol=1e-10)


@pytest.mark.gpu
def test_detsrm_gpu(seeded_rng, gpu_device):
    data, _, _ = _synthetic_subjects(seeded_rng)
    model = DetSRM(n_iter=10, features=5, rand_seed=0,
                   device="cuda").fit(data)
    for w in model.w_:
        assert np.allclose(w.T @ w, np.eye(5), atol=1e-4)
    projected = model.transform(data)
    c = np.corrcoef(projected[0].ravel(), projected[1].ravel())[0, 1]
    assert abs(c) > 0.9


def test_srm_loads_frozen_v0_1_model():
    """Backward-compat: the npz schema written by v0.1 keeps loading
    (the reference's sr_v0_4.npz compat test, ref
    tests/funcalign/test_srm.py)."""
    from pathlib import Path
    path = Path(__file__).parent / "data" / "srm_v0_1.npz"
    from brainiak_amd.funcalign.srm import load
    m = load(str(path))
    assert m.s_.shape == (5, 30)
    assert len(m.w_) == 3 and m.w_[0].shape == (40, 5)
    # transform with the loaded bases works
    rng = np.random.RandomState(7)
    S = rng.randn(5, 30)
    q, _ = np.linalg.qr(rng.randn(40, 5))
    x = q @ S
    proj = m.transform([x, None, None])
    assert proj[0].shape == (5, 30)
