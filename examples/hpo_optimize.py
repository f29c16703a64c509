#!/usr/bin/env python3
"""Hyperparameter optimization with the TPE-like sampler
(the reference's hyperparamopt example)."""

import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.hyperparamopt.hpo import fmin


def branin(args):
    x, y = args['x'], args['y']
    return ((y - 5.1 / (4 * np.pi ** 2) * x ** 2 + 5 * x / np.pi - 6) ** 2
            + 10 * (1 - 1 / (8 * np.pi)) * np.cos(x) + 10)


def main():
    import scipy.stats as st
    np.random.seed(0)
    space = {'x': {'dist': st.uniform(loc=-5, scale=15), 'lo': -5.,
                   'hi': 10.},
             'y': {'dist': st.uniform(loc=0, scale=15), 'lo': 0.,
                   'hi': 15.}}
    trials = []
    best = fmin(branin, space, max_evals=80, trials=trials,
                init_random_evals=20)
    print(f"best found: f({best['x']:.2f}, {best['y']:.2f}) = "
          f"{branin(best):.3f}  (global minimum 0.398)")


if __name__ == "__main__":
    main()
