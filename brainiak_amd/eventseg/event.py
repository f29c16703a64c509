"""Event segmentation: HMM with left-to-right event chains + sink state.

API parity with the reference (ref src/brainiak/eventseg/event.py:64-692):
annealed EM ``fit`` (variance schedule 4*0.98^(step-1)), Gaussian
log-observation model measuring Pearson-style distances on space-z-scored
data, ``find_events``/``predict``, ``set_event_patterns``, ``model_prior``,
``calc_weighted_event_var``, and split-merge proposals.

MI355X-first redesign (round 2): the reference runs a *per-dataset*
Python forward-backward in exp-space with per-step rescaling
(ref event.py:284-369, sequential ``for i in range(n_train)``).  Here the
whole E-step is a **batched pure-log-space recursion on torch**: one
``[B, T, K+1]`` tensor carries every dataset (and, during split-merge,
every proposal x dataset pair) through a single T-loop of batched
``logsumexp`` contractions, so the only sequential dimension is time and
everything else is device-parallel.  Log-space ``logsumexp`` replaces the
reference's exp/dot/rescale round trip — numerically it is at least as
stable and needs no masked-log inside the recursion (transition zeros
enter once as -inf in the log transition matrix).

The reference's Cython ``masked_log`` helper (eventseg/_utils.pyx:27-54)
survives as ``brainiak_amd.ops.masked_log`` for the public quirk surface
(prior vectors with exact zeros).

Citation context: [BaldassanoC2017] "Discovering event structure in
continuous narrative perception and memory", Neuron 95(3).
"""

import itertools
import logging

import numpy as np
import torch

from .. import ops

logger = logging.getLogger(__name__)

__all__ = ["EventSegment"]

_NEG_INF = float("-inf")


class NotFittedError(ValueError):
    pass


def _as_valid_2d(X):
    X = np.asarray(X, dtype=np.float64)
    if X.ndim != 2:
        raise ValueError("Input must be 2-D")
    if not np.all(np.isfinite(X)):
        raise ValueError("Input contains NaN or infinity")
    return X


def _zscore_rows(a, dim, ddof=1):
    """Torch z-score along ``dim`` (ddof matching scipy.stats.zscore)."""
    mu = a.mean(dim=dim, keepdim=True)
    sd = a.std(dim=dim, unbiased=(ddof == 1), keepdim=True)
    return (a - mu) / sd


class _ChainGraph:
    """Transition structure of the left-to-right chain(s) + sink state,
    held directly in log space.

    States 0..K-1 are events; state K is the absorbing sink.  Each chain
    starts with probability 1/n_chains, self-loops with 1 - p_trans and
    advances (or exits to the sink from its last event) with
    p_trans = (chain_len - 1) / T  — the reference's hazard choice
    (ref event.py:284-369), which makes the expected dwell time span the
    sequence.
    """

    def __init__(self, event_chains, t, device, dtype=torch.float64):
        chains = np.asarray(event_chains)
        k = chains.shape[0]
        labels = np.unique(chains, return_inverse=True)[1]
        n_chains = int(labels.max()) + 1

        start = torch.zeros(k + 1, dtype=dtype)
        end = torch.zeros(k + 1, dtype=dtype)
        P = torch.zeros((k + 1, k + 1), dtype=dtype)
        for c in range(n_chains):
            members = np.flatnonzero(labels == c)
            start[members[0]] = 1.0 / n_chains
            end[members[-1]] = 1.0 / n_chains
            p_trans = (len(members) - 1) / t
            if p_trans >= 1:
                raise ValueError("Too few timepoints")
            for j, s in enumerate(members):
                P[s, s] = 1 - p_trans
                nxt = members[j + 1] if j + 1 < len(members) else k
                P[s, nxt] = P[s, nxt] + p_trans
        P[k, k] = 1.0

        self.p_start = start.numpy()
        self.p_end = end.numpy()
        self.P = P.numpy()
        # log-space copies on the compute device (zeros -> -inf, the
        # masked_log contract, applied once up front)
        self.log_start = ops.masked_log(start).to(device)
        self.log_end = ops.masked_log(end).to(device)
        self.logP = ops.masked_log(P.reshape(-1)).reshape(k + 1,
                                                          k + 1).to(device)


class EventSegment:
    """HMM event segmentation (see module docstring).

    Parameters: ``n_events``, ``step_var`` (annealing schedule),
    ``n_iter``, ``event_chains``, ``split_merge``,
    ``split_merge_proposals``; plus ``device`` for the torch math.
    """

    @staticmethod
    def _default_var_schedule(step):
        return 4 * (0.98 ** (step - 1))

    def __init__(self, n_events=2, step_var=None, n_iter=500,
                 event_chains=None, split_merge=False,
                 split_merge_proposals=1, device="cpu"):
        self.n_events = n_events
        self.step_var = step_var if step_var is not None else \
            EventSegment._default_var_schedule
        self.n_iter = n_iter
        self.split_merge = split_merge
        self.split_merge_proposals = split_merge_proposals
        self.device = device
        self.event_chains = (np.zeros(n_events) if event_chains is None
                             else event_chains)

    # -- estimator params --------------------------------------------------

    def get_params(self, deep=True):
        return {"n_events": self.n_events, "step_var": self.step_var,
                "n_iter": self.n_iter, "event_chains": self.event_chains,
                "split_merge": self.split_merge,
                "split_merge_proposals": self.split_merge_proposals}

    def set_params(self, **params):
        for k, v in params.items():
            setattr(self, k, v)
        return self

    # -- batched core ------------------------------------------------------

    def _forward_backward_batch(self, logprob):
        """Batched log-space forward-backward.

        Parameters
        ----------
        logprob : torch [B, T, K] log observation probabilities (no sink
            column — it is appended here as -inf).

        Returns
        -------
        log_gamma : torch [B, T, K] row-normalized posteriors.
        ll : torch [B] sequence log-likelihoods.

        Everything is pure log space: one batched ``logsumexp``
        contraction against the [K+1, K+1] log transition matrix per
        timestep, no per-step rescaling needed (fp64 headroom is vast at
        fMRI sequence lengths).
        """
        B, T, K = logprob.shape
        dev = logprob.device
        graph = _ChainGraph(self.event_chains, T, dev)
        # compat surface: the reference exposes these after any FB pass
        self.p_start, self.p_end, self.P = (graph.p_start, graph.p_end,
                                            graph.P)

        obs = torch.full((B, T, K + 1), _NEG_INF, dtype=torch.float64,
                         device=dev)
        obs[:, :, :K] = logprob

        logP = graph.logP                     # [K+1, K+1]
        alphas = torch.empty((B, T, K + 1), dtype=torch.float64,
                             device=dev)
        a = graph.log_start[None, :] + obs[:, 0]
        alphas[:, 0] = a
        for i in range(1, T):
            # a[b, j] + logP[j, k] -> logsumexp over j
            a = torch.logsumexp(a[:, :, None] + logP[None], dim=1) \
                + obs[:, i]
            alphas[:, i] = a

        betas = torch.empty_like(alphas)
        b = graph.log_end[None, :].expand(B, K + 1).contiguous()
        betas[:, T - 1] = b
        for i in range(T - 2, -1, -1):
            w = b + obs[:, i + 1]             # [B, K+1]
            b = torch.logsumexp(logP[None] + w[:, None, :], dim=2)
            betas[:, i] = b

        log_gamma = alphas + betas
        log_gamma = log_gamma - torch.logsumexp(log_gamma, dim=2,
                                                keepdim=True)
        ll = torch.logsumexp(alphas[:, T - 1] + graph.log_end[None, :],
                             dim=1)
        return log_gamma[:, :, :K], ll

    def _forward_backward(self, logprob):
        """Single-sequence wrapper; returns (log_gamma [T, K], ll)."""
        lp = torch.as_tensor(np.asarray(logprob, dtype=np.float64),
                             device=torch.device(self.device))
        lg, ll = self._forward_backward_batch(lp[None])
        return lg[0].cpu().numpy(), float(ll[0])

    def _logprob_obs_batch(self, data, mean_pat, var):
        """[B, T, K] log N(d_t ; m_k, var_k I) on space-z-scored data,
        normalized by n_vox (the reference's per-voxel scaling).

        data : torch [B, V, T]; mean_pat : torch [V, K];
        var : scalar or [K].
        """
        B, V, T = data.shape
        K = mean_pat.shape[1]
        v = torch.as_tensor(
            np.broadcast_to(np.asarray(var, float), (K,)).copy(),
            dtype=torch.float64, device=data.device)
        dz = _zscore_rows(data, dim=1)        # z over voxels per column
        mz = _zscore_rows(mean_pat, dim=0)
        # ||d - m||^2 expanded so the cross term is one batched GEMM
        d2 = (dz * dz).sum(dim=1)             # [B, T]
        m2 = (mz * mz).sum(dim=0)             # [K]
        cross = torch.einsum("bvt,vk->btk", dz, mz)
        sq = d2[:, :, None] + m2[None, None, :] - 2.0 * cross
        lp = -0.5 * (V * torch.log(2 * torch.pi * v)[None, None, :]
                     + sq / v[None, None, :])
        return lp / V

    def _logprob_obs(self, data, mean_pat, var):
        """Single-dataset [T, K] observation log-probabilities (compat
        shim over the batched path)."""
        dev = torch.device(self.device)
        d = torch.as_tensor(np.asarray(data, dtype=np.float64), device=dev)
        m = torch.as_tensor(np.asarray(mean_pat, dtype=np.float64),
                            device=dev)
        return self._logprob_obs_batch(d[None], m, var)[0].cpu().numpy()

    # -- fitting -----------------------------------------------------------

    def _fit_validate(self, X):
        if len(np.unique(self.event_chains)) > 1:
            raise RuntimeError("Cannot fit chains, use set_event_patterns")
        if type(X) is not list:
            X = [X]
        dev = torch.device(self.device)
        out = []
        n_dim = None
        for x in X:
            x = _as_valid_2d(x).T              # [V, T]
            if n_dim is None:
                n_dim = x.shape[0]
            elif x.shape[0] != n_dim:
                raise ValueError("All datasets must share the voxel "
                                 "dimension")
            t = torch.as_tensor(x, dtype=torch.float64, device=dev)
            out.append(_zscore_rows(t, dim=1))  # z over time per voxel
        return out

    @staticmethod
    def _length_groups(X):
        """Group dataset indices by sequence length so ragged inputs
        still ride batched recursions (one batch per distinct T)."""
        groups = {}
        for i, x in enumerate(X):
            groups.setdefault(int(x.shape[1]), []).append(i)
        return groups

    def _e_step(self, X, mean_pat, var, groups):
        """Batched E-step over all datasets: returns per-dataset
        log_gamma list (torch) and ll [n] numpy."""
        n = len(X)
        log_gamma = [None] * n
        ll = np.empty(n)
        for _t, idxs in groups.items():
            batch = torch.stack([X[i] for i in idxs])     # [b, V, T]
            lp = self._logprob_obs_batch(batch, mean_pat, var)
            lg, l = self._forward_backward_batch(lp)
            for j, i in enumerate(idxs):
                log_gamma[i] = lg[j]
                ll[i] = float(l[j])
        return log_gamma, ll

    @staticmethod
    def _mean_pattern(X, log_gamma):
        """Average (over datasets) of X_b @ column-normalized gamma_b —
        the M-step event patterns, one batched contraction per dataset."""
        acc = None
        for x, lg in zip(X, log_gamma):
            g = torch.exp(lg)                  # [T, K]
            g = g / g.sum(dim=0, keepdim=True)
            p = x @ g                          # [V, K]
            acc = p if acc is None else acc + p
        return acc / len(X)

    def fit(self, X, y=None):
        """Anneal the event variance downward, alternating mean-pattern
        and segmentation updates until the log-likelihood decreases."""
        X = self._fit_validate(X)
        n_train = len(X)
        groups = self._length_groups(X)
        self.classes_ = np.arange(self.n_events)
        dev = torch.device(self.device)

        log_gamma = [torch.zeros((x.shape[1], self.n_events),
                                 dtype=torch.float64, device=dev)
                     for x in X]
        best_ll = float("-inf")
        self.ll_ = np.empty((0, n_train))
        for step in range(1, self.n_iter + 1):
            iteration_var = self.step_var(step)
            mean_pat = self._mean_pattern(X, log_gamma)
            log_gamma_new, ll = self._e_step(X, mean_pat, iteration_var,
                                             groups)

            if step > 1 and self.split_merge:
                ll, log_gamma_new, mean_pat = self._split_merge(
                    X, log_gamma_new, iteration_var, ll, groups)

            self.ll_ = np.vstack([self.ll_, ll[None, :]])
            if np.mean(ll) < best_ll:
                self.ll_ = self.ll_[:-1, :]
                break
            log_gamma = log_gamma_new
            self.segments_ = [torch.exp(lg).cpu().numpy()
                              for lg in log_gamma]
            self.event_var_ = iteration_var
            self.event_pat_ = mean_pat.cpu().numpy()
            best_ll = float(np.mean(ll))
            logger.debug("Fitting step %d, LL=%f", step, best_ll)
        return self

    # -- batched independent fits (MI355X addition) ------------------------

    def _logprob_obs_batch_perregion(self, data, mean_pat, var):
        """Like _logprob_obs_batch but with PER-REGION patterns:
        data [B, V, T], mean_pat [B, V, K] -> [B, T, K]."""
        B, V, T = data.shape
        K = mean_pat.shape[2]
        v = torch.as_tensor(
            np.broadcast_to(np.asarray(var, float), (K,)).copy(),
            dtype=torch.float64, device=data.device)
        dz = _zscore_rows(data, dim=1)
        mz = _zscore_rows(mean_pat, dim=1)
        d2 = (dz * dz).sum(dim=1)                      # [B, T]
        m2 = (mz * mz).sum(dim=1)                      # [B, K]
        cross = torch.bmm(dz.transpose(1, 2), mz)      # [B, T, K]
        sq = d2[:, :, None] + m2[:, None, :] - 2.0 * cross
        lp = -0.5 * (V * torch.log(2 * torch.pi * v)[None, None, :]
                     + sq / v[None, None, :])
        return lp / V

    def fit_regions(self, datasets):
        """Fit INDEPENDENT segmentations for many regions in one
        batched EM (MI355X addition — no reference counterpart).

        The per-region model is exactly ``fit`` on that region alone
        (same annealing schedule, same per-region early stop when the
        log-likelihood decreases); regions of equal (T, V) shape share
        each batched observation/forward-backward/M-step contraction,
        which is what makes many small searchlight-sized regions
        GPU-viable (one [B, T, K] recursion instead of B kernel-launch
        -bound fits).  split_merge is not supported here.

        Parameters
        ----------
        datasets : list of [T_r, V_r] arrays (TRs by voxels).

        Returns
        -------
        list of fitted EventSegment instances (segments_, event_pat_,
        ll_, event_var_ populated), one per region, in input order.
        """
        if self.split_merge:
            raise ValueError("fit_regions does not support split_merge")
        dev = torch.device(self.device)
        K = self.n_events
        models = [EventSegment(self.n_events, step_var=self.step_var,
                               n_iter=self.n_iter, device=self.device)
                  for _ in datasets]
        # group regions by (T, V) so each group is one dense batch
        groups = {}
        for i, d in enumerate(datasets):
            d2 = _as_valid_2d(d)
            groups.setdefault(d2.shape, []).append(i)
        for (T, V), idxs in groups.items():
            B = len(idxs)
            X = torch.stack([
                _zscore_rows(torch.as_tensor(
                    np.asarray(datasets[i], dtype=np.float64).T,
                    device=dev), dim=1)
                for i in idxs])                        # [B, V, T]
            log_gamma = torch.zeros((B, T, K), dtype=torch.float64,
                                    device=dev)
            best_ll = np.full(B, -np.inf)
            ll_hist = [[] for _ in range(B)]
            active = np.ones(B, dtype=bool)
            frozen_gamma = log_gamma.clone()
            frozen_pat = torch.zeros((B, V, K), dtype=torch.float64,
                                     device=dev)
            frozen_var = np.zeros(B)
            for step in range(1, self.n_iter + 1):
                if not active.any():
                    break
                iteration_var = self.step_var(step)
                g = torch.exp(log_gamma)
                g = g / g.sum(dim=1, keepdim=True).clamp_min(1e-300)
                mean_pat = torch.bmm(X, g)             # [B, V, K]
                lp = self._logprob_obs_batch_perregion(
                    X, mean_pat, iteration_var)
                lg_new, ll_t = self._forward_backward_batch(lp)
                ll = ll_t.cpu().numpy()
                improved = active & (ll >= best_ll)
                stopped = active & ~improved
                active = improved.copy()
                if improved.any():
                    m = torch.as_tensor(improved, device=dev)
                    log_gamma = torch.where(m[:, None, None], lg_new,
                                            log_gamma)
                    frozen_gamma = torch.where(m[:, None, None],
                                               lg_new, frozen_gamma)
                    frozen_pat = torch.where(m[:, None, None],
                                             mean_pat, frozen_pat)
                    frozen_var[improved] = iteration_var
                    best_ll[improved] = ll[improved]
                    for b in np.nonzero(improved)[0]:
                        ll_hist[b].append(ll[b])
                del stopped
            seg = torch.exp(frozen_gamma).cpu().numpy()
            pat = frozen_pat.cpu().numpy()
            for j, i in enumerate(idxs):
                mdl = models[i]
                mdl.classes_ = np.arange(K)
                mdl.segments_ = [seg[j]]
                mdl.event_pat_ = pat[j]
                mdl.event_var_ = frozen_var[j]
                mdl.ll_ = np.asarray(ll_hist[j])[:, None]
        return models

    def find_events_regions(self, models, datasets, var=None):
        """Batched inference counterpart of ``fit_regions``
        (MI355X addition): segment region r's dataset with region r's
        fitted model, all regions riding one [B, T, K] forward-
        backward per shape group.

        Parameters
        ----------
        models : list of fitted EventSegment (e.g. from fit_regions).
        datasets : list of [T_r, V_r] arrays, same length.
        var : optional shared variance override.

        Returns
        -------
        (segments, lls): list of [T_r, K] soft segmentations and an
        [n_regions] array of log-likelihoods, in input order.
        """
        if len(models) != len(datasets):
            raise ValueError("models and datasets must pair up")
        dev = torch.device(self.device)
        K = self.n_events
        groups = {}
        for i, d in enumerate(datasets):
            d2 = _as_valid_2d(d)
            groups.setdefault(d2.shape, []).append(i)
        segments = [None] * len(datasets)
        lls = np.empty(len(datasets))
        for (T, V), idxs in groups.items():
            X = torch.stack([
                torch.as_tensor(np.asarray(datasets[i],
                                           dtype=np.float64).T,
                                device=dev) for i in idxs])
            pats = torch.stack([
                torch.as_tensor(np.asarray(models[i].event_pat_,
                                           dtype=np.float64),
                                device=dev) for i in idxs])
            vs = var if var is not None else models[idxs[0]].event_var_
            lp = self._logprob_obs_batch_perregion(X, pats, vs)
            lg, ll = self._forward_backward_batch(lp)
            seg = torch.exp(lg).cpu().numpy()
            ll_h = ll.cpu().numpy()
            for j, i in enumerate(idxs):
                segments[i] = seg[j]
                lls[i] = ll_h[j]
        return segments, lls

    # -- inference ---------------------------------------------------------

    def set_event_patterns(self, event_pat):
        if event_pat.shape[1] != self.n_events:
            raise ValueError("Number of columns of event_pat must match "
                             "number of events")
        self.event_pat_ = event_pat.copy()

    def find_events(self, testing_data, var=None, scramble=False):
        """Segment a new dataset with the learned event patterns;
        returns (soft segmentation [T, K], log-likelihood)."""
        if var is None:
            if not hasattr(self, "event_var_"):
                raise NotFittedError("Event variance must be provided, if "
                                     "not previously set by fit()")
            var = self.event_var_
        if not hasattr(self, "event_pat_"):
            raise NotFittedError("The event patterns must first be set "
                                 "by fit() or set_event_patterns()")
        if scramble:
            mean_pat = self.event_pat_[:, np.random.permutation(
                self.n_events)]
        else:
            mean_pat = self.event_pat_
        logprob = self._logprob_obs(testing_data.T, mean_pat, var)
        lg, test_ll = self._forward_backward(logprob)
        return np.exp(lg), test_ll

    def predict(self, X):
        """Hard event label per timepoint (argmax of find_events)."""
        if not hasattr(self, "event_pat_") or \
                not hasattr(self, "event_var_"):
            raise NotFittedError("fit() has not been run")
        X = _as_valid_2d(X)
        segments, _ = self.find_events(X)
        return np.argmax(segments, axis=1)

    def calc_weighted_event_var(self, D, weights, event_pat):
        """Weighted variance of timepoints around each event pattern,
        with near-zero weights dropped (reference's 1/1000 floor)."""
        dev = torch.device(self.device)
        d = torch.as_tensor(np.asarray(D, dtype=np.float64), device=dev)
        dz = _zscore_rows(d, dim=1)            # [T, V], z over voxels
        w = torch.as_tensor(np.asarray(weights, dtype=np.float64),
                            device=dev)
        pat = torch.as_tensor(np.asarray(event_pat, dtype=np.float64),
                              device=dev)
        K = pat.shape[1]
        out = torch.empty(K, dtype=torch.float64)
        for e in range(K):
            keep = w[:, e] > w[:, e].max() / 1000
            we = w[keep, e]
            resid = dz[keep] - pat[:, e][None, :]
            sumsq = (we * (resid * resid).sum(dim=1)).sum()
            denom = we.sum() - (we * we).sum() / we.sum()
            out[e] = sumsq / denom
        return (out / D.shape[1]).cpu().numpy()

    def model_prior(self, t):
        """Prior segmentation (forward-backward with flat observations)."""
        lg, ll = self._forward_backward(np.zeros((t, self.n_events)))
        return np.exp(lg), ll

    # -- split-merge -------------------------------------------------------

    def _proposal_patterns(self, X, log_gamma):
        """Merge (adjacent-pair average) and split (half-mass) patterns.

        Returns torch ``merge_pat [V, K]`` and ``split_pat [V, 2K]``,
        averaged over datasets.  The half-split point of event e is the
        first timepoint where its cumulative soft mass reaches 0.5.
        """
        K = self.n_events
        merge_acc = split_acc = None
        for x, lg in zip(X, log_gamma):
            g = torch.exp(lg)
            g = g / g.sum(dim=0, keepdim=True)            # [T, K]
            T = g.shape[0]
            cs = torch.cumsum(g, dim=0)
            split_w = torch.zeros((T, 2 * K), dtype=g.dtype,
                                  device=g.device)
            for e in range(K):
                mid = int(torch.nonzero(cs[:, e] >= 0.5)[0])
                first_mass = float(cs[mid, e] - g[mid, e])
                split_w[:mid, 2 * e] = g[:mid, e] / first_mass
                split_w[mid:, 2 * e + 1] = g[mid:, e] / (1 - first_mass)
            # merge weights: average event e with its right neighbor
            merge_w = torch.zeros_like(g)
            merge_w[:, :K - 1] = 0.5 * (g[:, :K - 1] + g[:, 1:])
            merge_w[:, K - 1] = g[:, K - 1]
            mp = x @ merge_w
            sp = x @ split_w
            merge_acc = mp if merge_acc is None else merge_acc + mp
            split_acc = sp if split_acc is None else split_acc + sp
        return merge_acc / len(X), split_acc / len(X)

    @staticmethod
    def _columns_corr(a, b):
        """Pearson r between column a [V] and each column of b [V, m]."""
        az = a - a.mean()
        bz = b - b.mean(dim=0, keepdim=True)
        num = az @ bz
        den = az.norm() * bz.norm(dim=0)
        return num / den

    def _candidate_patterns(self, mean_pat, merge_pat, split_pat,
                            m_e, s_e):
        """Pattern matrix with event ``s_e`` split in two and events
        ``m_e, m_e+1`` merged (keeping K columns total)."""
        cols = []
        for e in range(self.n_events):
            if e == s_e:
                cols.append(split_pat[:, 2 * e])
                cols.append(split_pat[:, 2 * e + 1])
            elif e == m_e:
                cols.append(merge_pat[:, e])
            elif e == m_e + 1:
                continue
            else:
                cols.append(mean_pat[:, e])
        return torch.stack(cols, dim=1)

    def _split_merge(self, X, log_gamma, iteration_var, curr_ll, groups):
        """Propose event merges/splits to escape local minima; evaluate
        every (merge, split) candidate for every dataset in ONE batched
        forward-backward pass and keep the best improvement."""
        mean_pat = self._mean_pattern(X, log_gamma)
        merge_pat, split_pat = self._proposal_patterns(X, log_gamma)

        K = self.n_events
        split_corr = torch.empty(K)
        merge_corr = torch.empty(K)
        for e in range(K):
            split_corr[e] = self._columns_corr(
                mean_pat[:, e], split_pat[:, 2 * e:2 * e + 2]).max()
            merge_corr[e] = self._columns_corr(
                merge_pat[:, e], mean_pat[:, e:min(e + 2, K)]).min()
        merge_rank = torch.argsort(merge_corr[:K - 1],
                                   descending=True).tolist()
        split_rank = torch.argsort(split_corr).tolist()
        pairs = [(m, s) for m, s in itertools.product(
            merge_rank[:self.split_merge_proposals],
            split_rank[:self.split_merge_proposals])
            if s not in (m, m + 1)]
        if not pairs:
            return curr_ll, log_gamma, mean_pat

        candidates = [self._candidate_patterns(mean_pat, merge_pat,
                                               split_pat, m, s)
                      for m, s in pairs]

        best_ll, best_lg, best_mp = curr_ll, log_gamma, mean_pat
        best_mean = float(np.mean(curr_ll))
        n = len(X)
        for ci, cand in enumerate(candidates):
            # all datasets (per length group) through one batched pass
            lg_c, ll_c = self._e_step(X, cand, iteration_var, groups)
            if float(np.mean(ll_c)) > best_mean:
                best_mean = float(np.mean(ll_c))
                best_ll, best_lg, best_mp = ll_c, lg_c, cand
                m, s = pairs[ci]
                logger.debug("Identified merge %d,%d and split %d",
                             m, m + 1, s)
        del n
        return best_ll, best_lg, best_mp
