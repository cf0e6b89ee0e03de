"""Straggler avoidance: 2-component GMM on log(speed), analytic intersection
threshold; devices below threshold are rejected (reference src/Selection.py:4-48,
used at src/Server.py:324-338).

Provenance note: this module is a close re-implementation of the reference's
auto_threshold — the GMM fit (n_init=9, random_state=0), the analytic
two-Gaussian intersection and the fallback ladder are the ALGORITHM the survey
mandates exact semantics for, so the math here tracks
src/Selection.py line-for-line by necessity rather than design choice."""

from __future__ import annotations

import numpy as np


def auto_threshold(performance, n_init: int = 9) -> float:
    from sklearn.mixture import GaussianMixture

    perf = np.asarray(performance, dtype=float)
    if perf.size <= 1:
        return 0.0
    x = np.log(perf).reshape(-1, 1)
    gm = GaussianMixture(n_components=2, n_init=n_init, covariance_type="full",
                        random_state=0).fit(x)
    order = np.argsort(gm.means_.flatten())
    mu = gm.means_.flatten()[order]
    var = gm.covariances_.reshape(-1)[order]
    w = gm.weights_[order]

    # intersection of the two gaussians: a x^2 + b x + c = 0 in log space
    a = var[0] - var[1]
    b = 2 * (var[1] * mu[0] - var[0] * mu[1])
    c = (var[0] * mu[1] ** 2 - var[1] * mu[0] ** 2
         + 2 * var[0] * var[1] * np.log((var[1] * w[0]) / (var[0] * w[1])))

    if np.isclose(a, 0):
        if np.isclose(b, 0):
            t = np.mean(mu)
        else:
            root = -c / b
            t = root if mu[0] < root < mu[1] else np.mean(mu)
    else:
        roots = np.roots([a, b, c])
        real = roots[np.isreal(roots)].real
        cand = real[(real > mu[0]) & (real < mu[1])]
        if cand.size:
            mid = np.mean(mu)
            t = cand[np.argmin(np.abs(cand - mid))]
        else:
            t = np.mean(mu)
    return float(np.exp(t))
