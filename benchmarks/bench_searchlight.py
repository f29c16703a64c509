#!/usr/bin/env python3
"""BASELINE config 3: distributed Searchlight, whole-brain 91x109x91,
radius 3, over RCCL.  Blocks are sharded cyclically across ranks; each
rank runs a GPU-BATCHED block function (per-voxel inter-subject
correlation + ball aggregation as a conv3d) instead of the reference's
per-voxel Python loop — the MI355X-native way to run a numeric
searchlight.  Metric: searchlight centers evaluated per second."""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from benchmarks.common import dist_setup, emit, teardown, timed_steps  # noqa: E402


def _ball_kernel(rad, device):
    from brainiak_amd.searchlight import Ball
    mask = torch.as_tensor(Ball(rad).mask_.astype(np.float32),
                           device=device)
    return mask[None, None] / mask.sum()


def gpu_isc_block_fn(data, msk, rad, bcast_var, extra):
    """Mean within-ball inter-subject correlation for every center of a
    halo-padded block, batched on the device."""
    device = bcast_var
    a = torch.as_tensor(data[0], device=device)   # [x, y, z, T]
    b = torch.as_tensor(data[1], device=device)
    az = (a - a.mean(-1, keepdim=True))
    bz = (b - b.mean(-1, keepdim=True))
    denom = (az.norm(dim=-1) * bz.norm(dim=-1)).clamp_min(1e-12)
    corr = (az * bz).sum(-1) / denom              # [x, y, z]
    kernel = _ball_kernel(rad, device)
    ball_mean = torch.nn.functional.conv3d(
        corr[None, None], kernel)[0, 0]           # valid → inner block
    out = ball_mean.cpu().numpy()
    inner_msk = msk[rad:-rad, rad:-rad, rad:-rad] if rad > 0 else msk
    # float array with NaN at inactive centers (the stitcher accepts any
    # ndarray; avoids a per-center Python loop)
    return np.where(inner_msk, out, np.nan)


def gpu_isc_batch_fn(stacks, masks, rad, bcast_var, extra):
    """Batched device form for run_batched_block_function_device: the
    whole same-shape block group (already resident on device) rides
    ONE correlation + ONE ball aggregation (ops.stencil3d — direct LDS
    stencil instead of MIOpen's im2col conv lowering)."""
    from brainiak_amd import ops
    device = bcast_var
    a = (stacks[0] if torch.is_tensor(stacks[0])
         else torch.as_tensor(stacks[0], device=device)).float()
    b = (stacks[1] if torch.is_tensor(stacks[1])
         else torch.as_tensor(stacks[1], device=device)).float()
    az = a - a.mean(-1, keepdim=True)
    bz = b - b.mean(-1, keepdim=True)
    denom = (az.norm(dim=-1) * bz.norm(dim=-1)).clamp_min(1e-12)
    corr = (az * bz).sum(-1) / denom               # [B, x, y, z]
    kernel = _ball_kernel(rad, a.device)
    if a.is_cuda and ops.require_hip():
        ball_mean = ops.stencil3d(corr.contiguous(), kernel[0, 0])
    else:
        ball_mean = torch.nn.functional.conv3d(
            corr[:, None], kernel)[:, 0]           # [B, ox, oy, oz]
    inner = masks[:, rad:-rad, rad:-rad, rad:-rad] if rad > 0 else masks
    if torch.is_tensor(inner):
        return torch.where(inner, ball_mean,
                           torch.full_like(ball_mean, float("nan")))
    out = ball_mean.cpu().numpy()
    return np.where(inner, out, np.nan)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--dims", type=int, nargs=3, default=[91, 109, 91])
    ap.add_argument("--trs", type=int, default=40)
    ap.add_argument("--rad", type=int, default=3)
    ap.add_argument("--max-blk-edge", type=int, default=24)
    args = ap.parse_args()

    rank, world, device, _ = dist_setup()
    # let MIOpen pick a direct conv algorithm (the default heuristic
    # lands on im2col for the 7^3 ball kernel — 32 % of device time
    # in the profile)
    torch.backends.cudnn.benchmark = True
    from brainiak_amd.parallel import DistContext
    from brainiak_amd.searchlight import Ball, Searchlight
    ctx = DistContext(device=device)

    dims = tuple(args.dims) if device.type == "cuda" else (24, 24, 24)
    rng = np.random.RandomState(7)
    mask = np.ones(dims, dtype=bool)
    subjects = [rng.rand(*dims, args.trs).astype(np.float32)
                if rank == 0 else None for _ in range(2)]

    sl = Searchlight(sl_rad=args.rad, max_blk_edge=args.max_blk_edge,
                     shape=Ball, comm=ctx)
    sl.distribute(subjects, mask)
    sl.broadcast(device)

    def step(i):
        if device.type == "cuda":
            sl.run_batched_block_function_device(gpu_isc_batch_fn,
                                                 device)
        else:
            sl.run_batched_block_function(gpu_isc_batch_fn)

    elapsed = timed_steps(step, args.steps, args.warmup, world, device)
    centers = int(mask.sum())
    centers_per_sec = centers * args.steps / elapsed
    emit(rank, "searchlight_centers_per_sec", centers_per_sec,
         "centers/s", world, args.steps, args.warmup, elapsed, True,
         "strong", "fp32",
         {"model": "searchlight_isc", "dims": list(dims),
          "trs": args.trs, "rad": args.rad,
          "global_batch": centers, "seq_len": args.trs,
          "parallelism": f"block-sharded dp{world}"})
    teardown(world)


if __name__ == "__main__":
    main()
