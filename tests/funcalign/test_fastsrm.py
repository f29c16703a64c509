import numpy as np
import pytest

from brainiak_amd.funcalign.fastsrm import FastSRM
from brainiak_amd.funcalign.srm import NotFittedError


def _fastsrm_data(rng, subjects=4, voxels=80, trs=50, k=4, sessions=1):
    S = rng.randn(k, trs * sessions)
    data = []
    for _ in range(subjects):
        q, _ = np.linalg.qr(rng.randn(voxels, k))
        full = q @ S + 0.05 * rng.randn(voxels, trs * sessions)
        if sessions == 1:
            data.append(full)
        else:
            data.append([full[:, j * trs:(j + 1) * trs]
                         for j in range(sessions)])
    return data, S


def test_fastsrm_no_atlas(seeded_rng):
    data, S = _fastsrm_data(seeded_rng)
    model = FastSRM(atlas=None, n_components=4, n_iter=20, seed=0)
    model.fit(data)
    assert len(model.basis_list) == 4
    for b in model.basis_list:
        assert b.shape == (4, 80)
        assert np.allclose(b @ b.T, np.eye(4), atol=1e-8)
    shared = model.transform(data)
    assert shared.shape == (4, 50)
    # subjects should project to consistent shared responses
    per_subj = FastSRM(atlas=None, n_components=4, n_iter=20, seed=0,
                       aggregate=None).fit(data).transform(data)
    c = np.corrcoef(per_subj[0].ravel(), per_subj[1].ravel())[0, 1]
    assert abs(c) > 0.9


def test_fastsrm_prob_atlas(seeded_rng):
    data, _ = _fastsrm_data(seeded_rng, voxels=60)
    # random probabilistic atlas with 10 supervoxels
    atlas = np.abs(seeded_rng.rand(10, 60))
    model = FastSRM(atlas=atlas, n_components=4, n_iter=20, seed=0)
    shared = model.fit_transform(data)
    assert shared.shape == (4, 50)


def test_fastsrm_deterministic_atlas(seeded_rng):
    data, _ = _fastsrm_data(seeded_rng, voxels=60)
    labels = seeded_rng.randint(0, 11, 60)  # 0 = ignored
    model = FastSRM(atlas=labels, n_components=4, n_iter=20, seed=0)
    model.fit(data)
    assert model.basis_list[0].shape == (4, 60)


def test_fastsrm_sessions_and_paths(tmp_path, seeded_rng):
    data, _ = _fastsrm_data(seeded_rng, sessions=2)
    # save to disk and pass paths
    paths = []
    for i, subj in enumerate(data):
        row = []
        for j, sess in enumerate(subj):
            p = tmp_path / f"s{i}_{j}.npy"
            np.save(p, sess)
            row.append(str(p))
        paths.append(row)
    model = FastSRM(atlas=None, n_components=4, n_iter=15, seed=0,
                    temp_dir=str(tmp_path))
    model.fit(paths)
    assert isinstance(model.basis_list[0], str)
    shared = model.transform(paths)
    assert len(shared) == 2           # per-session shared responses
    assert shared[0].shape == (4, 50)
    recon = model.inverse_transform(shared, subjects_indexes=[0])
    assert recon[0][0].shape == (80, 50)
    model.clean()
    assert model.basis_list is None


def test_fastsrm_add_subjects(seeded_rng):
    data, _ = _fastsrm_data(seeded_rng, subjects=5)
    model = FastSRM(atlas=None, n_components=4, n_iter=20, seed=0)
    model.fit(data[:4])
    shared = model.transform(data[:4])
    model.add_subjects([data[4]], shared)
    assert len(model.basis_list) == 5
    b = model.basis_list[4]
    assert np.allclose(b @ b.T, np.eye(4), atol=1e-8)


def test_fastsrm_not_fitted(seeded_rng):
    data, _ = _fastsrm_data(seeded_rng)
    with pytest.raises(NotFittedError):
        FastSRM(n_components=4).transform(data)
    with pytest.raises(ValueError):
        FastSRM(aggregate="bogus")
