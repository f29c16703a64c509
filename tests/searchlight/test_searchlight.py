import numpy as np
import pytest

from brainiak_amd.parallel import spawn_ranks
from brainiak_amd.searchlight import Ball, Cube, Diamond, Searchlight


def test_shapes():
    assert Cube(1).mask_.sum() == 27
    assert Diamond(1).mask_.sum() == 7   # center + 6 neighbours
    assert Ball(1).mask_.sum() == 7
    assert Ball(2).mask_.shape == (5, 5, 5)
    d2 = Diamond(2).mask_
    assert d2[2, 2, 2] and d2[0, 2, 2] and not d2[0, 0, 0]


def _sum_fn(subjects, mask, rad, bcast):
    # sum of masked voxel values across the searchlight + bcast offset
    return float(sum(s[mask].sum() for s in subjects)) + bcast


def _run_serial(data, mask, rad=1, blk=3, bcast=0.0):
    sl = Searchlight(sl_rad=rad, max_blk_edge=blk)
    sl.distribute([data], mask)
    sl.broadcast(bcast)
    return sl.run_searchlight(_sum_fn, pool_size=1)


def test_searchlight_serial_correctness(seeded_rng):
    dim = (7, 8, 9)
    data = seeded_rng.rand(*dim, 4).astype(np.float32)
    mask = np.ones(dim, dtype=bool)
    out = _run_serial(data, mask, rad=1, blk=3, bcast=10.0)
    assert out.shape == dim
    # border trimmed
    assert out[0, 0, 0] is None
    # interior voxel: cube sum of 3^3 neighbourhood over all TRs + 10
    i, j, k = 3, 4, 5
    expected = data[i - 1:i + 2, j - 1:j + 2, k - 1:k + 2, :].sum() + 10.0
    assert np.isclose(out[i, j, k], expected, rtol=1e-5)


def test_searchlight_masked_voxels_skipped(seeded_rng):
    dim = (6, 6, 6)
    data = seeded_rng.rand(*dim, 3).astype(np.float32)
    mask = np.zeros(dim, dtype=bool)
    mask[2:4, 2:4, 2:4] = True
    out = _run_serial(data, mask, rad=1, blk=4)
    assert out[2, 2, 2] is not None
    assert out[1, 1, 1] is None
    # voxel_fn's mask argument excludes inactive voxels: compare against
    # explicitly masked sum
    i, j, k = 2, 3, 3
    region = data[i - 1:i + 2, j - 1:j + 2, k - 1:k + 2, :]
    m = mask[i - 1:i + 2, j - 1:j + 2, k - 1:k + 2]
    assert np.isclose(out[i, j, k], region[m].sum(), rtol=1e-5)


def test_searchlight_ball_shape_masks_voxel_fn(seeded_rng):
    dim = (5, 5, 5)
    data = seeded_rng.rand(*dim, 2).astype(np.float32)
    mask = np.ones(dim, dtype=bool)
    sl = Searchlight(sl_rad=1, max_blk_edge=5, shape=Diamond)
    sl.distribute([data], mask)
    sl.broadcast(0.0)
    out = sl.run_searchlight(_sum_fn, pool_size=1)
    i, j, k = 2, 2, 2
    region = data[i - 1:i + 2, j - 1:j + 2, k - 1:k + 2, :]
    expected = region[Diamond(1).mask_].sum()
    assert np.isclose(out[i, j, k], expected, rtol=1e-5)


def _dist_searchlight(ctx, outfile):
    rng = np.random.RandomState(5)
    dim = (9, 9, 9)
    data = rng.rand(*dim, 4).astype(np.float32)
    mask = np.ones(dim, dtype=bool)
    # subject owned by rank 0 only
    subjects = [data if ctx.rank == 0 else None]
    sl = Searchlight(sl_rad=1, max_blk_edge=3, comm=ctx)
    sl.distribute(subjects, mask)
    sl.broadcast(1.5)
    out = sl.run_searchlight(_sum_fn, pool_size=1)
    if ctx.rank == 0:
        vol = np.full(dim, np.nan)
        for idx in np.ndindex(dim):
            if out[idx] is not None:
                vol[idx] = out[idx]
        np.save(outfile, vol)


@pytest.mark.slow
def test_searchlight_distributed_matches_serial(tmp_path):
    out = str(tmp_path / "vol.npy")
    spawn_ranks(_dist_searchlight, world_size=2, args=(out,))
    vol_dist = np.load(out)

    rng = np.random.RandomState(5)
    dim = (9, 9, 9)
    data = rng.rand(*dim, 4).astype(np.float32)
    mask = np.ones(dim, dtype=bool)
    serial = _run_serial(data, mask, rad=1, blk=3, bcast=1.5)
    vol_serial = np.full(dim, np.nan)
    for idx in np.ndindex(dim):
        if serial[idx] is not None:
            vol_serial[idx] = serial[idx]
    assert np.allclose(vol_dist, vol_serial, equal_nan=True, rtol=1e-6)


def test_mvpa_voxelselector(seeded_rng):
    from sklearn import svm

    from brainiak_amd.fcma.mvpa_voxelselector import MVPAVoxelSelector
    dim = (6, 6, 6)
    n_epochs = 20
    labels = np.array([e % 2 for e in range(n_epochs)])
    data = seeded_rng.randn(*dim, n_epochs).astype(np.float32)
    # inject discriminative activity at one location
    data[3, 3, 3, :] += labels * 5.0
    mask = np.ones(dim, dtype=bool)
    sl = Searchlight(sl_rad=1, max_blk_edge=4, pool_size=1)
    mvs = MVPAVoxelSelector(data, mask, labels, 4, sl)
    vol, results = mvs.run(svm.SVC(kernel='linear'))
    assert len(results) == mask.sum()
    scores = [s for _, s in results]
    assert scores == sorted(scores, reverse=True)
    assert scores[0] > 0.8


def test_batched_block_function_matches_per_block(seeded_rng):
    """run_batched_block_function == run_block_function on the same
    numeric block fn."""
    dim = (10, 11, 9)
    data = [seeded_rng.rand(*dim, 6).astype(np.float32) for _ in range(2)]
    mask = seeded_rng.rand(*dim) > 0.2
    rad = 1

    def per_block(subjects, msk, r, bcast, extra=None):
        a, b = subjects
        s = (a.mean(-1) + b.mean(-1))
        inner = s[r:-r, r:-r, r:-r]
        im = msk[r:-r, r:-r, r:-r]
        return np.where(im, inner, np.nan)

    def batched(stacks, masks, r, bcast, extra=None):
        a, b = stacks
        s = a.mean(-1) + b.mean(-1)
        inner = s[:, r:-r, r:-r, r:-r]
        im = masks[:, r:-r, r:-r, r:-r]
        return np.where(im, inner, np.nan)

    sl1 = Searchlight(sl_rad=rad, max_blk_edge=4)
    sl1.distribute(data, mask)
    sl1.broadcast(None)
    out1 = sl1.run_block_function(per_block, pool_size=1)

    sl2 = Searchlight(sl_rad=rad, max_blk_edge=4)
    sl2.distribute(data, mask)
    sl2.broadcast(None)
    out2 = sl2.run_batched_block_function(batched)

    for i in range(dim[0]):
        for j in range(dim[1]):
            for k in range(dim[2]):
                v1, v2 = out1[i, j, k], out2[i, j, k]
                assert (v1 is None) == (v2 is None)
                if v1 is not None and not (np.isnan(v1) and np.isnan(v2)):
                    assert np.isclose(v1, v2, equal_nan=True)


def _batched_mean_fn(stacks, masks, rad, bcast, extra=None):
    s = stacks[0].mean(-1)[:, rad:-rad, rad:-rad, rad:-rad]
    im = masks[:, rad:-rad, rad:-rad, rad:-rad]
    return np.where(im, s + bcast, np.nan)


def _dist_batched_searchlight(ctx, outfile):
    rng = np.random.RandomState(6)
    dim = (9, 10, 8)
    data = rng.rand(*dim, 4).astype(np.float32)
    mask = rng.rand(*dim) > 0.25
    subjects = [data if ctx.rank == 0 else None]
    sl = Searchlight(sl_rad=1, max_blk_edge=4, comm=ctx)
    sl.distribute(subjects, mask)
    sl.broadcast(2.5)
    out = sl.run_batched_block_function(_batched_mean_fn)
    if ctx.rank == 0:
        vol = np.full(dim, np.nan)
        for idx in np.ndindex(dim):
            if out[idx] is not None:
                vol[idx] = out[idx]
        np.save(outfile, vol)


@pytest.mark.slow
def test_batched_block_distributed_matches_serial(tmp_path):
    out = str(tmp_path / "volb.npy")
    spawn_ranks(_dist_batched_searchlight, world_size=2, args=(out,))
    vol_dist = np.load(out)

    rng = np.random.RandomState(6)
    dim = (9, 10, 8)
    data = rng.rand(*dim, 4).astype(np.float32)
    mask = rng.rand(*dim) > 0.25
    sl = Searchlight(sl_rad=1, max_blk_edge=4)
    sl.distribute([data], mask)
    sl.broadcast(2.5)
    serial = sl.run_batched_block_function(_batched_mean_fn)
    vol_serial = np.full(dim, np.nan)
    for idx in np.ndindex(dim):
        if serial[idx] is not None:
            vol_serial[idx] = serial[idx]
    assert np.allclose(vol_dist, vol_serial, equal_nan=True, rtol=1e-6)


# -- round-2 depth (ref tests/searchlight/test_searchlight.py:1-307) --------

def _mean_sum_fn(subj, msk, rad, bcast):
    return float(np.sum([s.mean() for s in subj]))


def _token_fn(subj, msk, rad, bcast):
    return bcast["token"]


def _masksum_fn(subj, msk, rad, bcast):
    return int(msk.sum())


def _shape_check_fn(subj, msk, rad, bcast):
    assert subj[0].shape == tuple(bcast["expect"])
    return 1.0


def _one_fn(subj, msk, rad, bcast):
    return 1.0


def _depth_block_fn(subjects, msk, myrad, bcast, extra):
    out = np.empty((msk.shape[0] - 2, msk.shape[1] - 2,
                    msk.shape[2] - 2), dtype=object)
    out[:] = float(msk.shape[0])
    return out


def test_searchlight_rectangular_blocks(seeded_rng):
    """Non-cubic volume + block edge not dividing the volume: borders
    trimmed and stitched correctly."""
    dims = (9, 7, 11)
    data = [seeded_rng.rand(*dims, 3) for _ in range(2)]
    mask = np.ones(dims, dtype=bool)
    sl = Searchlight(sl_rad=1, max_blk_edge=4)
    sl.distribute(data, mask)
    sl.broadcast(None)
    out = sl.run_searchlight(_mean_sum_fn, pool_size=1)
    assert out.shape == dims
    # interior voxels computed; the rad-wide volume border is None
    # (the reference's block-trim semantics)
    assert all(out[i, j, k] is not None
               for i in range(1, dims[0] - 1)
               for j in range(1, dims[1] - 1)
               for k in range(1, dims[2] - 1))
    assert out[0, 0, 0] is None


def test_searchlight_bcast_var_reaches_fn(seeded_rng):
    dims = (5, 5, 5)
    data = [seeded_rng.rand(*dims, 2)]
    mask = np.ones(dims, dtype=bool)
    sl = Searchlight(sl_rad=1)
    sl.distribute(data, mask)
    sl.broadcast({"token": 17})
    out = sl.run_searchlight(_token_fn, pool_size=1)
    assert out[2, 2, 2] == 17


def test_searchlight_diamond_shape(seeded_rng):
    dims = (7, 7, 7)
    data = [seeded_rng.rand(*dims, 2)]
    mask = np.ones(dims, dtype=bool)
    sl = Searchlight(sl_rad=1, shape=Diamond)
    sl.distribute(data, mask)
    sl.broadcast(None)
    out = sl.run_searchlight(_masksum_fn, pool_size=1)
    assert out[3, 3, 3] == 7        # diamond: center + 6 neighbours


def test_searchlight_sparse_mask(seeded_rng):
    """Only masked voxels are computed; unmasked stay None; the window
    handed to voxel_fn is the full (2r+1)^3 sub-volume."""
    dims = (8, 8, 8)
    data = [seeded_rng.rand(*dims, 2)]
    mask = np.zeros(dims, dtype=bool)
    mask[3, 3, 3] = True
    mask[4, 5, 4] = True
    sl = Searchlight(sl_rad=2)
    sl.distribute(data, mask)
    sl.broadcast({"expect": (5, 5, 5, 2)})
    out = sl.run_searchlight(_shape_check_fn, pool_size=1)
    assert out[3, 3, 3] == 1.0 and out[4, 5, 4] == 1.0
    assert out[2, 2, 2] is None


def test_block_function_api(seeded_rng):
    """run_block_function hands the user raw halo blocks and stitches
    the trimmed outputs."""
    dims = (8, 8, 8)
    data = [seeded_rng.rand(*dims, 2)]
    mask = np.ones(dims, dtype=bool)
    sl = Searchlight(sl_rad=1, max_blk_edge=4)
    sl.distribute(data, mask)
    sl.broadcast(None)
    out = sl.run_block_function(_depth_block_fn, None, pool_size=1)
    assert out.shape == dims
    assert out[1, 1, 1] == 6.0      # 4 + 2*rad halo


def _identity_batch_fn(stacks, masks, rad, bcast, extra):
    import numpy as np
    a = stacks[0]
    if hasattr(a, "cpu"):
        a = a.cpu().numpy()
    # mean over time of the inner block
    out = a.mean(-1)[:, rad:-rad, rad:-rad, rad:-rad]
    return out


def test_device_batched_matches_host_batched(seeded_rng):
    """run_batched_block_function_device (cpu 'device') == the host
    path, including the resident-cache reuse on a second call."""
    dims = (10, 12, 10)
    data = [seeded_rng.rand(*dims, 5).astype(np.float32)]
    mask = np.ones(dims, dtype=bool)
    sl = Searchlight(sl_rad=1, max_blk_edge=5)
    sl.distribute(data, mask)
    sl.broadcast(None)
    host = sl.run_batched_block_function(_identity_batch_fn)
    dev1 = sl.run_batched_block_function_device(_identity_batch_fn,
                                                "cpu")
    dev2 = sl.run_batched_block_function_device(_identity_batch_fn,
                                                "cpu")   # cached path
    for a, b in ((host, dev1), (dev1, dev2)):
        mism = [(i, j, k) for i in range(dims[0])
                for j in range(dims[1]) for k in range(dims[2])
                if not np.isclose(float(a[i, j, k] or 0),
                                  float(b[i, j, k] or 0))]
        assert not mism, mism[:5]


def _count_mask_fn(d, m, r, b):
    return int(np.count_nonzero(m))


def test_min_active_voxels_proportion_gates_centers(seeded_rng):
    """Centers whose searchlight has too few in-mask voxels are left
    None when min_active_voxels_proportion is set."""
    dims = (7, 7, 7)
    data = [seeded_rng.rand(*dims, 4).astype(np.float32)]
    mask = np.zeros(dims, dtype=bool)
    mask[3, 3, 3] = True          # isolated center: sparse light
    mask[1, 1:6, 1:6] = True      # dense plane of centers
    sl_all = Searchlight(sl_rad=1, max_blk_edge=5)
    sl_all.distribute(data, mask)
    sl_all.broadcast(None)
    out_all = sl_all.run_searchlight(_count_mask_fn)
    assert out_all[3, 3, 3] == 1          # only itself in the light

    sl_gated = Searchlight(sl_rad=1, max_blk_edge=5,
                           min_active_voxels_proportion=0.2)
    sl_gated.distribute(data, mask)
    sl_gated.broadcast(None)
    out = sl_gated.run_searchlight(_count_mask_fn)
    # the isolated center falls below 20 % active and is skipped
    assert out[3, 3, 3] is None
    # dense-plane centers stay
    assert out[1, 3, 3] is not None
