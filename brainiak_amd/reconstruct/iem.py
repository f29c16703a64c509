"""Inverted encoding models (IEM) for 1-D circular and 2-D stimulus
spaces.

API parity with the reference (ref src/brainiak/reconstruct/iem.py:67-1050):
``InvertedEncoding1D`` (half-rectified exponentiated-sinusoid channel
basis; fit = pinv regression B = W·C; predict = channel inversion +
argmax; score = circular-distance R²) and ``InvertedEncoding2D``
(exponentiated 2-D cosine channels on square/triangular grids).

Citation: [Brouwer2009] "Decoding and reconstructing color from
responses in human visual cortex", J. Neurosci 29(44).
"""

import logging
import warnings

import numpy as np
import scipy.stats

from ..utils.utils import circ_dist

logger = logging.getLogger(__name__)

MAX_CONDITION_CHECK = 9000

__all__ = ["InvertedEncoding1D", "InvertedEncoding2D"]


class InvertedEncoding1D:
    """1-D circular/half-circular IEM; see module docstring."""

    def __init__(self, n_channels=6, channel_exp=5,
                 stimulus_mode='halfcircular', range_start=0.,
                 range_stop=180., channel_density=180,
                 stimulus_resolution=None):
        self.n_channels = n_channels
        self.channel_exp = channel_exp
        self.stimulus_mode = stimulus_mode
        self.range_start = range_start
        self.range_stop = range_stop
        self.channel_density = channel_density
        self.channel_domain = np.linspace(range_start, range_stop - 1,
                                          channel_density)
        self.stim_res = (channel_density if stimulus_resolution is None
                         else stimulus_resolution)
        self._check_params()

    def _check_params(self):
        if self.range_start >= self.range_stop:
            raise ValueError("range_start {} must be less than "
                             "{} range_stop.".format(self.range_start,
                                                     self.range_stop))
        if self.stimulus_mode == 'halfcircular':
            if (self.range_stop - self.range_start) != 180.:
                raise ValueError("For half-circular feature spaces,"
                                 "the range must be 180 degrees, "
                                 "not {}".format(self.range_stop
                                                 - self.range_start))
        elif self.stimulus_mode == 'circular':
            if (self.range_stop - self.range_start) != 360.:
                raise ValueError("For circular feature spaces, the"
                                 " range must be 360 degrees"
                                 "not {}".format(self.range_stop
                                                 - self.range_start))
        if self.n_channels < 2:
            raise ValueError("Insufficient number of channels.")
        if self.stimulus_mode not in ('circular', 'halfcircular'):
            raise ValueError("Stimulus mode must be one of these: "
                             "'circular', 'halfcircular'")

    def fit(self, X, y):
        """Estimate W from training data B = W·C."""
        if np.linalg.cond(X) > MAX_CONDITION_CHECK:
            logger.error("Data is singular.")
            raise ValueError("Data matrix is nearly singular.")
        if X.shape[0] < self.n_channels:
            raise ValueError("Fewer observations (trials) than "
                             "channels. Cannot compute pseudoinverse.")
        if np.ndim(X) != 2:
            raise ValueError("Data matrix has too many or too few "
                             "dimensions.")
        if np.shape(X)[0] != np.shape(y)[0]:
            raise ValueError("Mismatched data samples and label samples")

        self.channels_, channel_centers = self._define_channels()
        logger.info("Defined channels centered at %s degrees.",
                    np.rad2deg(channel_centers))
        C = self._define_trial_activations(y)
        self.W_ = X.transpose() @ np.linalg.pinv(C.transpose())
        if np.linalg.cond(self.W_) > MAX_CONDITION_CHECK:
            raise ValueError("Weight matrix is nearly singular.")
        return self

    def predict(self, X):
        """Predicted feature value per observation."""
        if np.ndim(X) != 2:
            raise ValueError("Data matrix has too many or too few "
                             "dimensions.")
        return self._predict_features(X)

    def score(self, X, y):
        """Circular-distance R² of predictions against y."""
        pred_features = self.predict(X)
        if self.stimulus_mode == 'halfcircular':
            pred_features = pred_features * 2
            y = y * 2
        ssres = (circ_dist(np.deg2rad(y),
                           np.deg2rad(pred_features)) ** 2).sum()
        sstot = (circ_dist(np.deg2rad(y),
                           np.ones(np.size(y)) * scipy.stats.circmean(
                               np.deg2rad(y))) ** 2).sum()
        return 1 - ssres / sstot

    def get_params(self, deep: bool = True):
        return {"n_channels": self.n_channels,
                "channel_exp": self.channel_exp,
                "stimulus_mode": self.stimulus_mode,
                "range_start": self.range_start,
                "range_stop": self.range_stop,
                "channel_domain": self.channel_domain,
                "stim_res": self.stim_res}

    def set_params(self, **parameters):
        for parameter, value in parameters.items():
            setattr(self, parameter, value)
        self.channel_domain = np.linspace(
            self.range_start, self.range_stop - 1, self.channel_density)
        self._check_params()
        return self

    def _define_channels(self):
        """Half-rectified sinusoid^exp channel basis."""
        channel_centers = np.linspace(np.deg2rad(self.range_start),
                                      np.deg2rad(self.range_stop),
                                      self.n_channels + 1)[:-1]
        if self.stimulus_mode == 'circular':
            domain = self.channel_domain * 0.5
            centers = channel_centers * 0.5
        else:
            domain = self.channel_domain
            centers = channel_centers
        channels = np.asarray(
            [np.cos(np.deg2rad(domain) - cx) ** self.channel_exp
             for cx in centers])
        return np.abs(channels), channel_centers

    def _define_trial_activations(self, stimuli):
        """[observations, n_channels] predicted channel responses."""
        stim_axis = np.linspace(self.range_start, self.range_stop - 1,
                                self.stim_res)
        stimuli = np.asarray(stimuli)
        if self.range_start > 0:
            stimuli = stimuli + self.range_start
        elif self.range_start < 0:
            stimuli = stimuli - self.range_start
        one_hot = np.eye(self.stim_res)
        indices = [np.argmin(abs(stim_axis - x)) for x in stimuli]
        stimulus_mask = one_hot[indices, :]
        if self.channel_density != self.stim_res:
            if self.channel_density % self.stim_res == 0:
                stimulus_mask = np.repeat(
                    stimulus_mask, self.channel_density // self.stim_res)
            else:
                raise NotImplementedError(
                    "Stimulus resolution must evenly divide the channel "
                    "density.")
        C = stimulus_mask @ self.channels_.transpose()
        if np.linalg.matrix_rank(C) < self.n_channels:
            warnings.warn("Stimulus matrix is {}, not full rank. May "
                          "cause issues with stimulus prediction/"
                          "reconstruction.".format(
                              np.linalg.matrix_rank(C)), RuntimeWarning)
        return C

    def _predict_channel_responses(self, X):
        return np.matmul(np.linalg.pinv(self.W_), X.transpose())

    def _predict_feature_responses(self, X):
        return np.matmul(self.channels_.transpose(),
                         self._predict_channel_responses(X))

    def _predict_features(self, X):
        pred_response = self._predict_feature_responses(X)
        feature_ind = np.argmax(pred_response, 0)
        return self.channel_domain[feature_ind]


class InvertedEncoding2D:
    """2-D IEM with exponentiated-cosine channels; see module docstring."""

    def __init__(self, stim_xlim, stim_ylim, stimulus_resolution,
                 stim_radius=None, chan_xlim=None, chan_ylim=None,
                 channels=None, channel_exp=7):
        if not isinstance(stimulus_resolution, list):
            stimulus_resolution = [stimulus_resolution,
                                   stimulus_resolution]
        if (len(stim_xlim) != 2) or (len(stim_ylim) != 2):
            raise ValueError(
                "Stimulus limits should be a sequence, 2 values")
        self.stim_fov = [stim_xlim, stim_ylim]
        self.stim_pixels = [np.linspace(stim_xlim[0], stim_xlim[1],
                                        stimulus_resolution[0]),
                            np.linspace(stim_ylim[0], stim_ylim[1],
                                        stimulus_resolution[1])]
        self.xp, self.yp = np.meshgrid(self.stim_pixels[0],
                                       self.stim_pixels[1])
        self.stim_radius_px = stim_radius
        self.channels = channels
        self.n_channels = None if channels is None else channels.shape[0]
        if chan_xlim is None:
            chan_xlim = stim_xlim
        if chan_ylim is None:
            chan_ylim = stim_ylim
        self.channel_limits = [chan_xlim, chan_ylim]
        self.channel_exp = channel_exp
        self._check_params()

    def _check_params(self):
        if len(self.stim_fov) != 2:
            raise ValueError(
                "Stim FOV needs to have an x-list and a y-list")
        if len(self.stim_fov[0]) != 2 or len(self.stim_fov[1]) != 2:
            raise ValueError(
                "Stimulus limits should be a sequence, 2 values")
        if (self.stim_fov[0][0] >= self.stim_fov[0][1]) or \
                (self.stim_fov[1][0] >= self.stim_fov[1][1]):
            raise ValueError(
                "Stimulus x or y limits should be ascending values")
        if self.xp.size != self.yp.size:
            raise ValueError("xpixel grid and ypixel grid do not have "
                             "same number of elements")
        if self.n_channels and np.all(self.channels):
            if self.n_channels != self.channels.shape[0]:
                raise ValueError(
                    "Number of channels {} does not match the defined "
                    "channels: {}".format(self.n_channels,
                                          self.channels.shape[0]))
            if self.channels.shape[1] != self.xp.size:
                raise ValueError(
                    "Defined {} channels over {} pixels, but stimuli are "
                    "represented over {} pixels. Pixels should match."
                    .format(self.n_channels, self.channels.shape[1],
                            self.xp.size))

    def fit(self, X, y, C=None):
        """Estimate W from training data; C defaults to circular-stimulus
        channel activations built from y."""
        if np.linalg.cond(X) > MAX_CONDITION_CHECK:
            raise ValueError("Data matrix is nearly singular.")
        if self.channels is None:
            raise ValueError(
                "Must define channels (set of basis functions).")
        if X.shape[0] < self.n_channels:
            raise ValueError("Fewer observations (trials) than "
                             "channels. Cannot compute pseudoinverse.")
        if np.shape(X)[0] != np.shape(y)[0]:
            raise ValueError("Mismatched data samples and label samples")
        if C is None:
            C = self._define_trial_activations(y)
        self.W_ = X.transpose() @ np.linalg.pinv(C.transpose())
        if np.linalg.cond(self.W_) > MAX_CONDITION_CHECK:
            raise ValueError("Weight matrix is nearly singular.")
        return self

    def predict(self, X):
        if np.ndim(X) != 2:
            raise ValueError("Data matrix has too many or too few "
                             "dimensions.")
        return self._predict_features(X)

    def score(self, X, y):
        """Per-observation R² of predicted 2-D features against y."""
        pred_features = self.predict(X)
        ssres = np.sum((pred_features - y) ** 2, axis=1)
        sstot = np.sum((y - np.mean(y)) ** 2, axis=1)
        return 1 - (ssres / sstot)

    def score_against_reconstructed(self, X, y, metric="euclidean"):
        """Distance between reconstructed pixel maps and expected maps."""
        from sklearn.metrics.pairwise import (
            cosine_distances,
            euclidean_distances,
        )
        yhat = self.predict_feature_responses(X)
        if metric == "euclidean":
            score_value = euclidean_distances(y.T, yhat.T)
        elif metric == "cosine":
            score_value = cosine_distances(y.T, yhat.T)
        else:
            raise ValueError("metric must be 'euclidean' or 'cosine'")
        return score_value[0, :]

    def get_params(self, deep: bool = True):
        return {"n_channels": self.n_channels,
                "channel_exp": self.channel_exp,
                "stim_fov": self.stim_fov,
                "stim_pixels": self.stim_pixels,
                "stim_radius_px": self.stim_radius_px, "xp": self.xp,
                "yp": self.yp, "channels": self.channels,
                "channel_limits": self.channel_limits}

    def set_params(self, **parameters):
        for parameter, value in parameters.items():
            setattr(self, parameter, value)
        self._check_params()
        return self

    def _make_2d_cosine(self, x, y, x_center, y_center, s):
        """Exponentiated 2-D cosine bumps (zero beyond radius s)."""
        cos_functions = np.zeros((len(x_center), len(x)))
        for i in range(len(x_center)):
            myr = np.sqrt((x - x_center[i]) ** 2
                          + (y - y_center[i]) ** 2).squeeze()
            qq = (myr <= s) * 1
            zp = (0.5 * (1 + np.cos(myr * np.pi / s))) ** self.channel_exp
            cos_functions[i, :] = zp * qq
        return cos_functions

    def _2d_cosine_sz_to_fwhm(self, size_constant):
        return 2 * size_constant * np.arccos(
            (0.5 ** (1 / self.channel_exp) - 0.5) / 0.5) / np.pi

    def _2d_cosine_fwhm_to_sz(self, fwhm):
        return (0.5 * np.pi * fwhm) / np.arccos(
            (0.5 ** (1 / self.channel_exp) - 0.5) / 0.5)

    def define_basis_functions_sqgrid(self, nchannels, channel_size=None):
        """Square-grid channel layout; sets self.channels."""
        if not isinstance(nchannels, list):
            nchannels = [nchannels, nchannels]
        chan_xcenters = np.linspace(self.channel_limits[0][0],
                                    self.channel_limits[0][1],
                                    nchannels[0])
        chan_ycenters = np.linspace(self.channel_limits[1][0],
                                    self.channel_limits[1][1],
                                    nchannels[1])
        cx, cy = np.meshgrid(chan_xcenters, chan_ycenters)
        cx = cx.reshape(-1, 1)
        cy = cy.reshape(-1, 1)
        if channel_size is None:
            channel_size = 1.2 * (chan_xcenters[1] - chan_xcenters[0])
        cos_width = self._2d_cosine_fwhm_to_sz(channel_size)
        self.channels = self._make_2d_cosine(
            self.xp.reshape(-1, 1), self.yp.reshape(-1, 1), cx, cy,
            cos_width)
        self.n_channels = self.channels.shape[0]
        return self.channels, np.hstack([cx, cy])

    def define_basis_functions_trigrid(self, grid_radius,
                                       channel_size=None):
        """Triangular-grid channel layout; sets self.channels."""
        x_dist = np.diff(self.channel_limits[0]) / (grid_radius * 2)
        y_dist = x_dist * np.sqrt(3) * 0.5
        trigrid = np.zeros((0, 2))
        xbase = np.expand_dims(
            np.arange(self.channel_limits[0][0],
                      self.channel_limits[0][1], x_dist.item()), 1)
        for yi, yv in enumerate(np.arange(self.channel_limits[1][0],
                                          self.channel_limits[1][1],
                                          y_dist.item())):
            if (yi % 2) == 0:
                xx = xbase.copy()
            else:
                xx = xbase.copy() + x_dist / 2
            yy = np.ones((xx.size, 1)) * yv
            trigrid = np.vstack((trigrid, np.hstack((xx, yy))))
        if channel_size is None:
            channel_size = 1.1 * x_dist
        cos_width = self._2d_cosine_fwhm_to_sz(channel_size)
        self.channels = self._make_2d_cosine(
            self.xp.reshape(-1, 1), self.yp.reshape(-1, 1),
            trigrid[:, 0], trigrid[:, 1], cos_width)
        self.n_channels = self.channels.shape[0]
        return self.channels, trigrid

    def _define_trial_activations(self, stim_centers, stim_radius=None):
        """[observations, channels] responses for circular stimuli."""
        nstim = stim_centers.shape[0]
        if self.stim_radius_px is None:
            if stim_radius is None:
                raise ValueError("No defined stimulus radius. Please set.")
            self.stim_radius_px = stim_radius
        if not isinstance(self.stim_radius_px, (np.ndarray, list)):
            self.stim_radius_px = np.ones(nstim) * self.stim_radius_px
        stimulus_mask = np.zeros((self.xp.size, nstim))
        for i in range(nstim):
            rad_vals = ((self.xp.reshape(-1, 1) - stim_centers[i, 0]) ** 2
                        + (self.yp.reshape(-1, 1)
                           - stim_centers[i, 1]) ** 2)
            inds = np.where(rad_vals < self.stim_radius_px[i])[0]
            stimulus_mask[inds, i] = 1
        C = self.channels.squeeze() @ stimulus_mask
        C = C.transpose()
        if np.linalg.matrix_rank(C) < self.n_channels:
            warnings.warn("Stimulus matrix is {}, not full rank. May "
                          "cause issues with stimulus prediction/"
                          "reconstruction.".format(
                              np.linalg.matrix_rank(C)), RuntimeWarning)
        return C

    def _predict_channel_responses(self, X):
        return np.matmul(np.linalg.pinv(self.W_), X.transpose())

    def predict_feature_responses(self, X):
        return np.matmul(self.channels.transpose(),
                         self._predict_channel_responses(X))

    def _predict_features(self, X):
        pred_response = self.predict_feature_responses(X)
        feature_ind = np.argmax(pred_response, 0)
        return np.hstack((self.xp.reshape(-1, 1)[feature_ind],
                          self.yp.reshape(-1, 1)[feature_ind]))
