#!/usr/bin/env python3
"""BASELINE config 4: ISFC leave-one-out, 50k voxels x 32 subjects,
subject-parallel over RCCL/xGMI.  One step = one full distributed ISFC
(summary_statistic='mean'): ONE all-reduce of the across-subject sum,
local [V, V] gemms per owned subject, one all-reduce of the arctanh
accumulator.  Metric: ISFC voxel-pair values per second (V^2 * subjects
/ time)."""

import argparse
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from benchmarks.common import dist_setup, emit, teardown, timed_steps  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--num-voxels", type=int, default=50000)
    ap.add_argument("--subjects", type=int, default=32)
    ap.add_argument("--trs", type=int, default=200)
    args = ap.parse_args()

    rank, world, device, _ = dist_setup()
    from brainiak_amd.isc import isfc_distributed
    from brainiak_amd.parallel import DistContext
    ctx = DistContext(device=device)

    # on CPU smoke runs shrink the problem so it finishes in seconds
    V = args.num_voxels if device.type == "cuda" else 512
    subjects = args.subjects if device.type == "cuda" else 4

    g = torch.Generator().manual_seed(1234 + rank)
    mine = [torch.randn((args.trs, V), generator=g).to(device)
            for s in range(subjects) if s % world == rank]

    precision = 'bf16' if device.type == "cuda" else 'fp32'

    def step(i):
        # device-resident result (the [V,V] D2H would dominate; GPU
        # consumers keep it in HBM — isc.py return_tensor)
        isfc_distributed(mine, ctx, summary_statistic='mean',
                         precision=precision,
                         return_tensor=device.type == "cuda")

    elapsed = timed_steps(step, args.steps, args.warmup, world, device)
    pairs_per_sec = float(V) * V * subjects * args.steps / elapsed
    emit(rank, "isfc_voxel_pairs_per_sec", pairs_per_sec, "pairs/s",
         world, args.steps, args.warmup, elapsed, True, "strong",
         precision, {"model": "isfc_leave_one_out", "num_voxels": V,
                  "subjects": subjects, "trs": args.trs,
                  "global_batch": subjects, "seq_len": args.trs,
                  "parallelism": f"subject-sharded dp{world}"})
    teardown(world)


if __name__ == "__main__":
    main()
