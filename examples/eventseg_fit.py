#!/usr/bin/env python3
"""Event segmentation (HMM) on synthetic event-structured data
(the reference's eventseg example)."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.eventseg import EventSegment


def main():
    rng = np.random.RandomState(0)
    n_events, event_len, n_vox = 5, 20, 30
    patterns = rng.randn(n_events, n_vox) * 2
    data = np.vstack([np.tile(p, (event_len, 1)) for p in patterns])
    data += 0.8 * rng.randn(*data.shape)

    es = EventSegment(n_events=n_events, n_iter=100)
    es.fit(data)
    bounds = np.where(np.diff(np.argmax(es.segments_[0], axis=1)))[0] + 1
    print("true boundaries:", [event_len * i for i in range(1, n_events)])
    print("found boundaries:", bounds.tolist())

    # segment new data with the learned event patterns
    test = np.vstack([np.tile(p, (event_len, 1)) for p in patterns])
    test += 0.8 * rng.randn(*test.shape)
    segments, ll = es.find_events(test)
    print(f"held-out log-likelihood: {ll:.1f}")


if __name__ == "__main__":
    main()
