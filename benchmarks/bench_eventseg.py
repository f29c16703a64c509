#!/usr/bin/env python3
"""Event segmentation throughput: batched HMM forward-backward over
many searchlight-sized regions (the realistic whole-brain use: fit an
EventSegment per region).  Metric: region-fits/s.

One step = fitting ``--regions`` independent EventSegment models
(ragged lengths grouped by the batched forward-backward)."""

import argparse
import os
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from benchmarks.common import dist_setup, emit, teardown, timed_steps  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--regions", type=int, default=64)
    ap.add_argument("--trs", type=int, default=200)
    ap.add_argument("--voxels", type=int, default=100)
    ap.add_argument("--events", type=int, default=8)
    args = ap.parse_args()

    rank, world, device, _ = dist_setup()
    from brainiak_amd.eventseg.event import EventSegment

    rng = np.random.RandomState(7 + rank)
    regions = args.regions if device.type == "cuda" else 8
    data = []
    for _ in range(regions):
        bounds = np.sort(rng.choice(
            np.arange(1, args.trs), args.events - 1, replace=False))
        means = rng.randn(args.events, args.voxels)
        seg = np.zeros((args.trs, args.voxels))
        prev = 0
        for e, b in enumerate(list(bounds) + [args.trs]):
            seg[prev:b] = means[e]
            prev = b
        data.append(seg + 0.5 * rng.randn(args.trs, args.voxels))

    dev = str(device) if device.type == "cuda" else "cpu"

    batched = not os.environ.get("BRAINIAK_EVENTSEG_SEQ")

    def step(i):
        if batched:
            EventSegment(args.events, n_iter=10,
                         device=dev).fit_regions(data)
        else:
            for d in data:
                EventSegment(args.events, n_iter=10,
                             device=dev).fit(d.copy())

    elapsed = timed_steps(step, args.steps, args.warmup, world, device)
    fits_per_sec = regions * world * args.steps / elapsed
    emit(rank, "eventseg_region_fits_per_sec", fits_per_sec, "fits/s",
         world, args.steps, args.warmup, elapsed, True, "weak", "fp64",
         {"model": "eventseg", "regions": regions, "trs": args.trs,
          "voxels": args.voxels, "events": args.events,
          "global_batch": regions, "seq_len": args.trs,
          "parallelism": f"region-sharded dp{world}"})
    teardown()


if __name__ == "__main__":
    main()
