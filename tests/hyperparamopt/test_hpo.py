import numpy as np
import scipy.stats as st

from brainiak_amd.hyperparamopt.hpo import (
    fmin,
    get_next_sample,
    get_sigma,
    gmm_1d_distribution,
)


def test_get_sigma():
    x = np.array([1.0, 2.0, 5.0])
    s = get_sigma(x, min_limit=0.0, max_limit=10.0)
    # point 1: left neighbor 0 (limit), right 2 → max(1, 1) = 1
    assert np.isclose(s[0], 1.0)
    # point 5: left 2, right 10 → max(3, 5) = 5
    assert np.isclose(s[2], 5.0)


def test_gmm_pdf_properties(seeded_rng):
    x = np.array([0.2, 0.5, 0.8])
    g = gmm_1d_distribution(x, min_limit=0.0, max_limit=1.0)
    assert g(-0.5) == 0 and g(1.5) == 0
    assert g(0.5) > 0
    # finite positive mass over the domain (exact normalization is not a
    # property of the reference's truncated-component weighting either)
    grid = np.linspace(0, 1, 400)
    total = np.trapz(g(grid), grid)
    assert 0.3 < total <= 1.2
    samples = g.get_samples(200)
    assert np.all((samples >= 0) & (samples <= 1))


def test_get_next_sample_prefers_good_region(seeded_rng):
    np.random.seed(0)
    x = np.concatenate([np.random.uniform(0, 0.3, 20),
                        np.random.uniform(0.7, 1.0, 20)])
    # loss low near x≈0.15
    y = (x - 0.15) ** 2
    nxt = get_next_sample(x, y, min_limit=0.0, max_limit=1.0)
    assert 0.0 <= nxt <= 1.0
    assert nxt < 0.55  # should lean toward the good region


def test_fmin_quadratic(seeded_rng):
    np.random.seed(3)
    trials = []
    space = {'x': {'dist': st.uniform(loc=0, scale=10), 'lo': 0,
                   'hi': 10}}
    best = fmin(lambda d: (d['x'] - 3.0) ** 2, space, max_evals=60,
                trials=trials, init_random_evals=20)
    assert len(trials) == 60
    assert abs(best['x'] - 3.0) < 1.0
    assert best['loss'] == min(t['loss'] for t in trials)


def test_fmin_validates_dist():
    import pytest
    with pytest.raises(ValueError):
        fmin(lambda d: 0.0, {'x': {'dist': 42}}, 1, [])


def test_gmm_matches_reference_oracles(seeded_rng):
    """The reference's gmm_1d_distribution oracles: density ordering,
    hard bounds, weight equivalence, scalar==array evaluation."""
    x = np.array([1., 1., 2., 3., 1.])
    d = gmm_1d_distribution(x, min_limit=0., max_limit=4.)
    assert d(1.1) > d(3.5)
    assert d(2.0) > d(3.0)
    assert d(-1.0) == 0 and d(9.0) == 0
    samples = d.get_samples(n=25)
    assert np.all(samples < 4.) and np.all(samples > 0.)

    # duplicate points == integer weights
    x_dup = np.array([1., 1., 2., 3., 1., 3.])
    d_dup = gmm_1d_distribution(x_dup)
    d_w = gmm_1d_distribution(np.array([1., 2., 3.]),
                              weights=np.array([3., 1., 2.]))
    y = d_w(np.array([1.1, 2.0]))
    assert d_w(1.1) == y[0]
    assert abs(d_dup(1.1) - d_w(1.1)) < 1e-5
    assert abs(d_dup(2.0) - d_w(2.0)) < 1e-5


def test_fmin_continues_trials(seeded_rng):
    """fmin resumes from an existing trials list (the reference's
    continuation contract) and improves or keeps the best loss."""
    def f(args):
        return args['x'] ** 2

    space = {'x': {'dist': st.uniform(loc=-10., scale=20),
                   'lo': -10., 'hi': 10.}}
    trials = []
    best1 = fmin(loss_fn=f, space=space, max_evals=30, trials=trials)
    n1 = len(trials)
    best2 = fmin(loss_fn=f, space=space, max_evals=30, trials=trials)
    assert len(trials) == n1 + 30
    assert best2['loss'] <= best1['loss']
    assert abs(best2['x']) < 2.5
