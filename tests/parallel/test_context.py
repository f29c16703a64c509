import numpy as np
import pytest
import torch

from brainiak_amd.parallel import DistContext, shard_slices, spawn_ranks


def test_shard_slices():
    s = shard_slices(10, 3)
    assert [x.stop - x.start for x in s] == [4, 3, 3]
    assert s[0].start == 0 and s[-1].stop == 10
    # more shards than items
    s = shard_slices(2, 4)
    assert [x.stop - x.start for x in s] == [1, 1, 0, 0]


def test_serial_context_noop_collectives():
    ctx = DistContext(device="cpu")
    assert ctx.world_size == 1 and ctx.rank == 0
    a = np.arange(4.0)
    assert np.allclose(ctx.all_reduce(a), a)
    assert ctx.all_gather(a) == [a]
    assert ctx.broadcast_object({"x": 1}) == {"x": 1}
    assert ctx.all_gather_object(3) == [3]
    xs = ctx.all_reduce_many([a, np.float64(2.0) * np.ones(1)])
    assert np.allclose(xs[0], a)
    ctx.barrier()  # no-op


def _collective_roundtrip(ctx, outfile):
    t = torch.full((3,), float(ctx.rank + 1))
    summed = ctx.all_reduce(t.clone())
    gathered = ctx.all_gather(np.full(2, ctx.rank, dtype=np.float64))
    obj = ctx.broadcast_object({"from": ctx.rank} if ctx.rank == 0 else None)
    fused = ctx.all_reduce_many(
        [np.ones(3) * ctx.rank, np.array([float(ctx.rank)])])
    if ctx.rank == 0:
        np.savez(outfile,
                 summed=summed.numpy(),
                 gathered=np.stack(gathered),
                 bcast=obj["from"],
                 fused0=fused[0], fused1=fused[1])


@pytest.mark.slow
def test_collectives_world2(tmp_path):
    out = str(tmp_path / "out.npz")
    spawn_ranks(_collective_roundtrip, world_size=2, args=(out,))
    z = np.load(out)
    assert np.allclose(z["summed"], 3.0)        # 1 + 2
    assert np.allclose(z["gathered"][0], 0.0)
    assert np.allclose(z["gathered"][1], 1.0)
    assert z["bcast"] == 0
    assert np.allclose(z["fused0"], 1.0)        # 0 + 1
    assert np.allclose(z["fused1"], 1.0)


# -- round-2 depth -----------------------------------------------------------

def _collectives_entry(ctx, outfile):
    import numpy as np
    # all_reduce on numpy
    x = np.full(4, float(ctx.rank + 1))
    summed = ctx.all_reduce(x, op="sum")
    # fused many: mixed shapes/dtypes ride one collective
    a = np.arange(3, dtype=np.float32) * (ctx.rank + 1)
    b = np.array([ctx.rank], dtype=np.int64)
    fa, fb = ctx.all_reduce_many([a, b])
    # broadcast object from rank 1
    obj = {"tag": ctx.rank} if ctx.rank == 1 else None
    got = ctx.broadcast_object(obj, src=1)
    # gather_object to root
    gathered = ctx.gather_object(ctx.rank * 10, dst=0)
    # scatter from root
    part = ctx.scatter_object(
        [f"part{i}" for i in range(ctx.world_size)]
        if ctx.is_root else None)
    if ctx.is_root:
        np.save(outfile, np.array([
            summed[0], fa[1], fb[0], got["tag"],
            gathered[1], float(part == "part0")]))


def test_collectives_two_ranks(tmp_path):
    from brainiak_amd.parallel import spawn_ranks
    out = str(tmp_path / "coll.npy")
    spawn_ranks(_collectives_entry, world_size=2, args=(out,))
    import numpy as np
    vals = np.load(out)
    assert vals[0] == 3.0          # 1 + 2
    assert vals[1] == 3.0          # 1*1 + 1*2 fused
    assert vals[2] == 1.0          # 0 + 1 int64 kept
    assert vals[3] == 1            # broadcast from rank 1
    assert vals[4] == 10           # gathered rank 1's value
    assert vals[5] == 1.0          # scatter delivered part0 to root


def test_serial_context_noop_semantics():
    import numpy as np
    from brainiak_amd.parallel import DistContext
    ctx = DistContext(device="cpu")
    assert not ctx.is_distributed and ctx.is_root
    x = np.arange(5.0)
    assert ctx.all_reduce(x) is x
    assert ctx.all_gather(x) == [x]
    assert ctx.broadcast_object({"a": 1}) == {"a": 1}
    assert ctx.gather_object(7) == [7]
    assert ctx.scatter_object([3]) == 3
    ctx.barrier()                  # no-op, must not hang
    assert ctx.shard(10) == slice(0, 10)
    assert ctx.owner_of(5, 10) == 0


def test_shard_slices_cover_and_balance():
    from brainiak_amd.parallel import shard_slices
    for n, k in ((10, 3), (7, 7), (5, 8), (100, 8)):
        sl = shard_slices(n, k)
        covered = []
        for s in sl:
            covered.extend(range(*s.indices(n)))
        assert covered == list(range(n))
        sizes = [s.stop - s.start for s in sl]
        assert max(sizes) - min(sizes) <= 1


def _ws4_roundtrip(ctx, out_dir):
    import numpy as np

    assert ctx.world_size == 4
    # all-reduce across 4 ranks
    v = ctx.all_reduce(np.array([float(ctx.rank + 1)]), op="sum")
    assert float(v[0]) == 10.0
    # shard covers the range exactly once
    sl = ctx.shard(103)
    spans = ctx.all_gather_object((sl.start, sl.stop))
    covered = sorted(spans)
    assert covered[0][0] == 0 and covered[-1][1] == 103
    for (a, b), (c, d) in zip(covered, covered[1:]):
        assert b == c
    # broadcast from root
    obj = ctx.broadcast_object({"k": 7} if ctx.is_root else None)
    assert obj == {"k": 7}
    if ctx.is_root:
        (out_dir / "ok").write_text("1")


def test_collectives_world4(tmp_path):
    """4-rank gloo plumbing (the 8-GPU driver bench shape, scaled to
    what CPU CI can run)."""
    spawn_ranks(_ws4_roundtrip, world_size=4, args=(tmp_path,))
    assert (tmp_path / "ok").exists()
