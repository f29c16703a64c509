import numpy as np
import pytest

from brainiak_amd.eventseg.event import EventSegment, NotFittedError


def _event_data(rng, n_events=4, event_len=15, n_voxels=20, noise=0.5):
    """Piecewise-constant event patterns + noise."""
    patterns = rng.randn(n_events, n_voxels) * 2
    rows = []
    bounds = []
    for e in range(n_events):
        rows.append(np.tile(patterns[e], (event_len, 1)))
        bounds.append(e * event_len)
    data = np.vstack(rows) + noise * rng.randn(n_events * event_len,
                                               n_voxels)
    return data, np.array(bounds[1:]), patterns


def test_fit_recovers_event_boundaries(seeded_rng):
    data, bounds, _ = _event_data(seeded_rng)
    es = EventSegment(n_events=4, n_iter=100)
    es.fit(data)
    assert es.event_pat_.shape == (20, 4)
    assert len(es.segments_) == 1
    seg = es.segments_[0]
    assert seg.shape == (60, 4)
    # each timepoint's soft assignment sums to 1
    assert np.allclose(seg.sum(axis=1), 1.0, atol=1e-6)
    # hard segmentation should match the true boundaries closely
    labels = np.argmax(seg, axis=1)
    true_labels = np.repeat(np.arange(4), 15)
    assert np.mean(labels == true_labels) > 0.85
    # log-likelihood increased during fitting
    assert es.ll_.shape[0] >= 2
    assert np.mean(es.ll_[-1]) >= np.mean(es.ll_[0])


def test_find_events_on_new_data(seeded_rng):
    data, _, patterns = _event_data(seeded_rng)
    es = EventSegment(n_events=4, n_iter=60).fit(data)
    data2, _, _ = _event_data(np.random.RandomState(1))
    # transfer the learned patterns via set_event_patterns instead
    es2 = EventSegment(n_events=4)
    es2.set_event_patterns(es.event_pat_)
    seg, ll = es2.find_events(data, var=1.0)
    assert seg.shape == (60, 4)
    assert np.isfinite(ll)
    labels = np.argmax(seg, axis=1)
    true_labels = np.repeat(np.arange(4), 15)
    assert np.mean(labels == true_labels) > 0.8


def test_predict_and_notfitted(seeded_rng):
    data, _, _ = _event_data(seeded_rng)
    es = EventSegment(n_events=4, n_iter=60)
    with pytest.raises(NotFittedError):
        es.predict(data)
    es.fit(data)
    labels = es.predict(data)
    assert labels.shape == (60,)
    assert set(labels) <= set(range(4))
    # monotonically nondecreasing labels (left-to-right chain)
    assert np.all(np.diff(labels) >= 0)


def test_multiple_datasets(seeded_rng):
    d1, _, _ = _event_data(seeded_rng)
    d2, _, _ = _event_data(seeded_rng)
    es = EventSegment(n_events=4, n_iter=60).fit([d1, d2])
    assert len(es.segments_) == 2
    assert es.ll_.shape[1] == 2


def test_model_prior():
    es = EventSegment(n_events=3)
    seg, ll = es.model_prior(30)
    assert seg.shape == (30, 3)
    assert np.allclose(seg.sum(axis=1), 1.0)
    # prior probability of event 0 decreasing over time, event K-1
    # increasing
    assert seg[0, 0] > seg[-1, 0]
    assert seg[-1, 2] > seg[0, 2]


def test_split_merge_runs(seeded_rng):
    data, _, _ = _event_data(seeded_rng, n_events=3, event_len=10)
    es = EventSegment(n_events=3, n_iter=30, split_merge=True)
    es.fit(data)
    assert hasattr(es, "event_pat_")


def test_event_chains():
    es = EventSegment(n_events=4, event_chains=np.array([0, 0, 1, 1]))
    with pytest.raises(RuntimeError):
        es.fit(np.random.randn(40, 10))
    # with patterns set, find_events works on chains
    es.set_event_patterns(np.random.randn(10, 4))
    seg, ll = es.find_events(np.random.randn(40, 10), var=1.0)
    assert seg.shape == (40, 4)


def test_calc_weighted_event_var(seeded_rng):
    data, _, patterns = _event_data(seeded_rng, noise=0.1)
    es = EventSegment(n_events=4, n_iter=80).fit(data)
    ev_var = es.calc_weighted_event_var(data, es.segments_[0],
                                        es.event_pat_)
    assert ev_var.shape == (4,)
    assert np.all(ev_var >= 0)


# -- round-2 depth: reference assertion parity (ref tests/eventseg/
#    test_event.py:1-203) ------------------------------------------------

def test_ragged_multiple_datasets(seeded_rng):
    """Datasets of different lengths batch by length group and fit."""
    d1, _, _ = _event_data(seeded_rng, event_len=15)
    d2, _, _ = _event_data(seeded_rng, event_len=12)
    d3, _, _ = _event_data(seeded_rng, event_len=15)
    es = EventSegment(n_events=4, n_iter=60).fit([d1, d2, d3])
    assert len(es.segments_) == 3
    assert es.segments_[0].shape == (60, 4)
    assert es.segments_[1].shape == (48, 4)
    for seg in es.segments_:
        assert np.allclose(seg.sum(axis=1), 1.0, atol=1e-6)


def test_find_events_scramble_lowers_ll(seeded_rng):
    data, _, _ = _event_data(seeded_rng)
    es = EventSegment(n_events=4, n_iter=60).fit(data)
    np.random.seed(0)
    _, ll_true = es.find_events(data)
    lls = []
    for _ in range(5):
        _, ll_s = es.find_events(data, scramble=True)
        lls.append(ll_s)
    assert ll_true >= max(lls)


def test_fit_rejects_bad_inputs(seeded_rng):
    es = EventSegment(n_events=3)
    with pytest.raises(ValueError):
        es.fit(np.random.randn(20, 5, 2))     # 3-D
    bad = np.random.randn(30, 5)
    bad[3, 2] = np.nan
    with pytest.raises(ValueError):
        es.fit(bad)
    # mismatched voxel dimensions across datasets
    with pytest.raises(ValueError):
        es.fit([np.random.randn(30, 5), np.random.randn(30, 6)])


def test_too_few_timepoints_raises():
    es = EventSegment(n_events=10)
    with pytest.raises(ValueError):
        es.model_prior(5)     # p_trans = 9/5 >= 1


def test_event_transition_matrix_properties():
    es = EventSegment(n_events=4)
    es.model_prior(40)
    # set as side effect of any FB pass (reference behaviour)
    P = es.P
    assert P.shape == (5, 5)
    assert np.allclose(P.sum(axis=1), 1.0)
    # left-to-right: no backward probability
    assert np.all(np.tril(P, k=-1) == 0)
    assert es.p_start[0] == 1.0 and es.p_end[-2] == 1.0


def test_weighted_var_decreases_with_noise(seeded_rng):
    clean, _, _ = _event_data(np.random.RandomState(42), noise=0.02)
    noisy, _, _ = _event_data(np.random.RandomState(1), noise=1.5)
    es_c = EventSegment(n_events=4, n_iter=60).fit(clean)
    es_n = EventSegment(n_events=4, n_iter=60).fit(noisy)
    v_c = es_c.calc_weighted_event_var(clean, es_c.segments_[0],
                                       es_c.event_pat_)
    v_n = es_n.calc_weighted_event_var(noisy, es_n.segments_[0],
                                       es_n.event_pat_)
    assert np.mean(v_c) < np.mean(v_n)


def test_split_merge_improves_bad_init(seeded_rng):
    """Split-merge must not hurt: final LL with proposals >= without."""
    data, _, _ = _event_data(seeded_rng, n_events=5, event_len=8)
    plain = EventSegment(n_events=5, n_iter=40).fit(data.copy())
    sm = EventSegment(n_events=5, n_iter=40, split_merge=True,
                      split_merge_proposals=2).fit(data.copy())
    assert np.mean(sm.ll_[-1]) >= np.mean(plain.ll_[-1]) - 1e-6


def test_logprob_normalization_constant(seeded_rng):
    """_logprob_obs scales by 1/n_vox (reference quirk: per-voxel
    average log-likelihood)."""
    es = EventSegment(n_events=3)
    d = seeded_rng.randn(8, 10)     # [V, T] = 8 voxels, 10 TRs
    pat = seeded_rng.randn(8, 3)
    lp1 = es._logprob_obs(d, pat, 1.0)
    assert lp1.shape == (10, 3)
    # doubling the variance raises logprob of far points
    lp2 = es._logprob_obs(d, pat, 4.0)
    assert np.all(np.isfinite(lp1)) and np.all(np.isfinite(lp2))


def test_fit_regions_matches_individual_fits(seeded_rng):
    """Batched independent-region fit == per-region fit (same EM,
    shared recursions; deterministic given data)."""
    rng = seeded_rng
    K, T, V = 4, 60, 30
    regions = []
    for _ in range(5):
        bounds = np.sort(rng.choice(np.arange(1, T), K - 1,
                                    replace=False))
        means = rng.randn(K, V)
        seg = np.zeros((T, V))
        prev = 0
        for e, b in enumerate(list(bounds) + [T]):
            seg[prev:b] = means[e]
            prev = b
        regions.append(seg + 0.3 * rng.randn(T, V))

    batched = EventSegment(K, n_iter=25).fit_regions(regions)
    for d, mb in zip(regions, batched):
        ms = EventSegment(K, n_iter=25).fit(d.copy())
        assert np.allclose(mb.segments_[0], ms.segments_[0],
                           atol=1e-6)
        assert np.allclose(mb.event_pat_, ms.event_pat_, atol=1e-6)
        assert np.isclose(mb.ll_[-1, 0], ms.ll_[-1].mean(), atol=1e-8)
        # boundaries usable downstream
        assert mb.segments_[0].shape == (T, K)


def test_fit_regions_ragged_shapes(seeded_rng):
    """Different (T, V) regions batch by shape group and come back in
    input order."""
    rng = seeded_rng
    shapes = [(40, 20), (60, 25), (40, 20)]
    regions = [rng.randn(t, v) for t, v in shapes]
    models = EventSegment(3, n_iter=5).fit_regions(regions)
    for (t, v), m in zip(shapes, models):
        assert m.segments_[0].shape == (t, 3)

    with pytest.raises(ValueError):
        EventSegment(3, split_merge=True).fit_regions(regions)


def test_find_events_regions_matches_individual(seeded_rng):
    """Batched inference == per-region find_events."""
    rng = seeded_rng
    K, T, V = 3, 50, 20
    regions = [rng.randn(T, V) for _ in range(4)]
    models = EventSegment(K, n_iter=10).fit_regions(regions)
    test_sets = [rng.randn(T, V) for _ in range(4)]
    es = EventSegment(K, n_iter=10)
    segs, lls = es.find_events_regions(models, test_sets)
    for m, d, seg, ll in zip(models, test_sets, segs, lls):
        ref_seg, ref_ll = m.find_events(d)
        assert np.allclose(seg, ref_seg, atol=1e-9)
        assert np.isclose(ll, ref_ll, atol=1e-7)
    with pytest.raises(ValueError):
        es.find_events_regions(models, test_sets[:2])
