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
