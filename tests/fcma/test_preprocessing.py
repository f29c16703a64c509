"""Tests for fcma.preprocessing (ref tests/fcma/test_preprocessing.py)."""

import numpy as np

from brainiak_amd.fcma.preprocessing import (RandomType,
                                             generate_epochs_info,
                                             prepare_fcma_data,
                                             prepare_mvpa_data)
from brainiak_amd.nifti import NiftiImage


def _setup(rng, n_subj=2, trs=24, dims=(4, 4, 4)):
    imgs = [NiftiImage(rng.rand(*dims, trs), np.eye(4))
            for _ in range(n_subj)]
    mask = np.zeros(dims, dtype=bool)
    mask[1:3, 1:3, 1:3] = True
    # 2 conditions x 2 epochs of 6 TRs per subject
    cond = np.zeros((2, 4, trs), dtype=int)
    cond[0, 0, 0:6] = 1
    cond[1, 1, 6:12] = 1
    cond[0, 2, 12:18] = 1
    cond[1, 3, 18:24] = 1
    conditions = [cond] * n_subj
    return imgs, mask, conditions


def test_prepare_fcma_data(seeded_rng):
    imgs, mask, conditions = _setup(seeded_rng)
    raw1, raw2, labels = prepare_fcma_data(imgs, conditions, mask)
    assert raw2 is None
    assert len(raw1) == 8              # 4 epochs x 2 subjects
    assert labels == [0, 0, 1, 1] * 2   # condition-major per subject
    assert raw1[0].shape == (6, int(mask.sum()))
    # epochs are z-scored over time and 1/sqrt(len)-scaled
    col = raw1[0][:, 0]
    assert np.isclose(col.mean(), 0, atol=1e-5)
    assert np.isclose(np.sum(col ** 2), 1.0, atol=1e-4)


def test_prepare_fcma_data_two_masks(seeded_rng):
    imgs, mask, conditions = _setup(seeded_rng)
    mask2 = np.zeros(mask.shape, dtype=bool)
    mask2[0, 0, 0:2] = True
    raw1, raw2, labels = prepare_fcma_data(imgs, conditions, mask,
                                           mask2)
    assert raw2 is not None and raw2[0].shape == (6, 2)
    assert len(raw1) == len(raw2) == 8


def test_prepare_fcma_data_randomized(seeded_rng):
    imgs, mask, conditions = _setup(seeded_rng)
    r1, _, _ = prepare_fcma_data(imgs, conditions, mask)
    r1p, _, _ = prepare_fcma_data(imgs, conditions, mask,
                                  random=RandomType.REPRODUCIBLE)
    # voxel permutation: different arrangement, same multiset of
    # per-voxel column norms
    assert not np.allclose(r1[0], r1p[0])
    assert np.allclose(sorted(np.sum(r1[0] ** 2, axis=0)),
                       sorted(np.sum(r1p[0] ** 2, axis=0)), atol=1e-5)


def test_generate_epochs_info(seeded_rng):
    _, _, conditions = _setup(seeded_rng, n_subj=1)
    info = generate_epochs_info(conditions[:1])
    assert info == [(0, 0, 0, 6), (0, 0, 12, 18),
                    (1, 0, 6, 12), (1, 0, 18, 24)]


def test_prepare_mvpa_data(seeded_rng):
    imgs, mask, conditions = _setup(seeded_rng)
    processed, labels = prepare_mvpa_data(imgs, conditions, mask)
    assert processed.shape == (int(mask.sum()), 8)
    assert labels.tolist() == [0.0, 0.0, 1.0, 1.0] * 2
    # within-subject z-scoring across epochs
    assert np.allclose(processed[:, :4].mean(axis=1), 0, atol=1e-5)


def test_stage_timer_logs_and_profiles(tmp_path, caplog, monkeypatch):
    """stage_timer logs a duration line and (with
    BRAINIAK_TORCH_PROFILE) drops a chrome trace."""
    import logging

    from brainiak_amd.utils.timing import stage_timer
    log = logging.getLogger("timing_test")
    with caplog.at_level(logging.INFO, logger="timing_test"):
        with stage_timer("unit stage", log):
            pass
    assert any("unit stage took" in r.message for r in caplog.records)

    monkeypatch.setenv("BRAINIAK_TORCH_PROFILE", str(tmp_path))
    with stage_timer("profiled stage", log):
        import torch
        torch.ones(4) @ torch.ones(4)
    traces = list(tmp_path.glob("profiled_stage_*.json"))
    assert len(traces) == 1 and traces[0].stat().st_size > 0
