#!/usr/bin/env python3
"""Batched event segmentation over many regions — the MI355X-native
`fit_regions` API (one [B, T, K] forward-backward for all regions).

    python examples/eventseg_regions.py
"""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.eventseg.event import EventSegment


def main():
    rng = np.random.RandomState(0)
    K, T, V, n_regions = 5, 120, 40, 24
    regions, truth = [], []
    for _ in range(n_regions):
        bounds = np.sort(rng.choice(np.arange(1, T), K - 1,
                                    replace=False))
        means = rng.randn(K, V)
        seg = np.zeros((T, V))
        prev = 0
        for e, b in enumerate(list(bounds) + [T]):
            seg[prev:b] = means[e]
            prev = b
        regions.append(seg + 0.4 * rng.randn(T, V))
        truth.append(bounds)

    models = EventSegment(K, n_iter=40).fit_regions(regions)
    hits = 0
    for m, bounds in zip(models, truth):
        est = np.where(np.diff(np.argmax(m.segments_[0], axis=1)))[0] + 1
        hits += sum(min(abs(est - b)) <= 3 for b in bounds)
    print("recovered %d/%d event boundaries within 3 TRs across %d "
          "regions" % (hits, (K - 1) * n_regions, n_regions))


if __name__ == "__main__":
    main()
