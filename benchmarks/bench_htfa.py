#!/usr/bin/env python3
"""BASELINE config 5: HTFA, 100k voxels x 20 subjects, subjects sharded
across GPUs with the global-template all-reduce/broadcast per global
iteration (sized for 288 GB HBM: full-resolution subjects stay
resident).  One step = one HTFA fit with one global iteration.
Metric: subject-voxels processed per second."""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from benchmarks.common import dist_setup, emit, teardown, timed_steps  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--num-voxels", type=int, default=100000)
    ap.add_argument("--subjects", type=int, default=20)
    ap.add_argument("--trs", type=int, default=100)
    ap.add_argument("--K", type=int, default=20)
    args = ap.parse_args()

    rank, world, device, _ = dist_setup()
    from brainiak_amd.factoranalysis.htfa import HTFA
    from brainiak_amd.parallel import DistContext
    ctx = DistContext(device=device)

    V = args.num_voxels if device.type == "cuda" else 2000
    subjects = args.subjects if device.type == "cuda" else 4

    rng = np.random.RandomState(100 + rank)
    X, R = [], []
    for s in range(subjects):
        if s % world != rank:
            continue
        coords = rng.rand(V, 3) * 40
        centers = rng.rand(args.K, 3) * 40
        d2 = ((coords[:, None, :] - centers[None, :, :]) ** 2).sum(-1)
        F = np.exp(-d2 / 50.0)
        W = rng.randn(args.K, args.trs)
        X.append((F @ W + 0.1 * rng.randn(V, args.trs)))
        R.append(coords)

    def step(i):
        htfa = HTFA(K=args.K, n_subj=subjects, max_global_iter=1,
                    max_local_iter=1, comm=ctx,
                    device=str(device) if device.type == "cuda" else "cpu")
        htfa.fit(X, R)

    elapsed = timed_steps(step, args.steps, args.warmup, world, device)
    voxels_per_sec = float(V) * subjects * args.steps / elapsed
    emit(rank, "htfa_subject_voxels_per_sec", voxels_per_sec,
         "subject-voxels/s", world, args.steps, args.warmup, elapsed,
         True, "strong", "fp64",
         {"model": "htfa", "num_voxels": V, "subjects": subjects,
          "trs": args.trs, "K": args.K, "global_batch": subjects,
          "seq_len": args.trs,
          "parallelism": f"subject-sharded dp{world}"})
    teardown(world)


if __name__ == "__main__":
    main()
