"""TPE-style hyperparameter optimization (API parity: ref
src/brainiak/hyperparamopt/hpo.py:40-374).

``fmin`` fits two 1-D Gaussian mixtures per variable — one over the
best ~15% of trials (weighted by loss), one over the rest — and samples
the candidate maximizing the likelihood ratio (expected improvement),
with an exploration probability for random draws.

Citation: [Bergstra2013] "Making a science of model search", ICML 2013.
"""

import logging
import math

import numpy as np
import scipy.stats as st
from scipy.special import erf

logger = logging.getLogger(__name__)

__all__ = ["fmin"]


def get_sigma(x, min_limit=-np.inf, max_limit=np.inf):
    """Per-point GMM bandwidths: max distance to nearest neighbor (with
    the limits appended as virtual neighbors)."""
    z = np.append(x, [min_limit, max_limit])
    sigma = np.ones(x.shape)
    for i in range(x.size):
        left = z[z < x[i]]
        right = z[z > x[i]]
        xleft = left.max() if left.size else -np.inf
        xright = right.min() if right.size else np.inf
        sigma[i] = max(x[i] - xleft, xright - x[i])
        if sigma[i] == np.inf:
            sigma[i] = min(x[i] - xleft, xright - x[i])
        if sigma[i] == -np.inf:  # pragma: no cover - degenerate
            sigma[i] = 1.0
    return sigma


class gmm_1d_distribution:
    """Weighted 1-D GMM over a set of points, truncated to
    [min_limit, max_limit]; callable for pdf, ``get_samples`` to draw."""

    def __init__(self, x, min_limit=-np.inf, max_limit=np.inf,
                 weights=1.0):
        self.points = x
        self.N = x.size
        self.min_limit = min_limit
        self.max_limit = max_limit
        self.sigma = get_sigma(x, min_limit=min_limit, max_limit=max_limit)
        # renormalize each truncated component to unit mass
        self.weights = (2 / (erf((max_limit - x)
                                 / (np.sqrt(2.) * self.sigma))
                             - erf((min_limit - x)
                                   / (np.sqrt(2.) * self.sigma)))
                        * weights)
        self.W_sum = np.sum(self.weights)

    def get_gmm_pdf(self, x):
        if x < self.min_limit or x > self.max_limit:
            return 0
        y = 0.0
        for i in range(self.points.size):
            z = (x - self.points[i]) / self.sigma[i]
            y += (math.exp(-0.5 * z * z)
                  / (math.sqrt(2. * np.pi) * self.sigma[i])
                  * self.weights[i]) / self.W_sum
        return y

    def __call__(self, x):
        if np.isscalar(x):
            return self.get_gmm_pdf(x)
        return np.array([self.get_gmm_pdf(t) for t in x])

    def get_samples(self, n):
        normalized_w = self.weights / np.sum(self.weights)
        idx = st.rv_discrete(values=(range(self.N),
                                     normalized_w)).rvs(size=max(n, 1))
        samples = np.zeros(n)
        k = j = 0
        while k < n:
            i = idx[j]
            j += 1
            if j == len(idx):
                idx = st.rv_discrete(values=(range(self.N),
                                             normalized_w)).rvs(size=n)
                j = 0
            v = np.random.normal(loc=self.points[i], scale=self.sigma[i])
            if self.min_limit <= v <= self.max_limit:
                samples[k] = v
                k += 1
        return samples


def get_next_sample(x, y, min_limit=-np.inf, max_limit=np.inf):
    """Candidate with the best EI ratio l(x)/g(x) between the good-trial
    and rest-trial GMMs, avoiding near-duplicates of past samples."""
    z = np.array(list(zip(x, y)),
                 dtype=np.dtype([('x', float), ('y', float)]))
    z = np.sort(z, order='y')
    n = y.shape[0]
    g = int(np.round(np.ceil(0.15 * n)))
    ldata = z[0:g]
    gdata = z[g:n]
    lymin = ldata['y'].min()
    lymax = ldata['y'].max()
    if lymax > lymin:
        weights = (lymax - ldata['y']) / (lymax - lymin)
    else:
        weights = np.ones(ldata['x'].size)
    lx = gmm_1d_distribution(ldata['x'], min_limit=min_limit,
                             max_limit=max_limit, weights=weights)
    gx = gmm_1d_distribution(gdata['x'], min_limit=min_limit,
                             max_limit=max_limit)

    samples = lx.get_samples(n=1000)
    ei = lx(samples) / np.maximum(gx(samples), 1e-300)

    h = (x.max() - x.min()) / (10 * x.size)
    s = 0
    while np.abs(x - samples[ei.argmax()]).min() < h:
        ei[ei.argmax()] = 0
        s += 1
        if s == samples.size:
            break
    return samples[ei.argmax()]


def fmin(loss_fn, space, max_evals, trials, init_random_evals=30,
         explore_prob=0.2):
    """Minimize ``loss_fn`` over the hyperparameter ``space``.

    space example: ``{'x': {'dist': scipy.stats.uniform(0, 1),
    'lo': 0, 'hi': 1}}``.  Appends each trial (dict of values + 'loss')
    to ``trials`` and returns the best one.
    """
    for s in space:
        if not hasattr(space[s]['dist'], 'rvs'):
            raise ValueError('Unknown distribution type for variable')
        if 'lo' not in space[s]:
            space[s]['lo'] = -np.inf
        if 'hi' not in space[s]:
            space[s]['hi'] = np.inf

    if len(trials) > init_random_evals:
        init_random_evals = 0

    for t in range(max_evals):
        sdict = {}
        use_random_sampling = not (t >= init_random_evals
                                   and np.random.random() > explore_prob)
        yarray = np.array([tr['loss'] for tr in trials])
        for s in space:
            sarray = np.array([tr[s] for tr in trials])
            if use_random_sampling:
                sdict[s] = space[s]['dist'].rvs()
            else:
                sdict[s] = get_next_sample(sarray, yarray,
                                           min_limit=space[s]['lo'],
                                           max_limit=space[s]['hi'])
        logger.debug('Explore' if use_random_sampling else 'Exploit')
        y = loss_fn(sdict)
        sdict['loss'] = y
        trials.append(sdict)

    yarray = np.array([tr['loss'] for tr in trials])
    return trials[yarray.argmin()]
