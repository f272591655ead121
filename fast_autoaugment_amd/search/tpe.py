"""Tree-structured Parzen Estimator search (hyperopt-TPE semantics).

Replaces the reference's HyperOptSearch-over-Ray-Tune (search.py:216-245).
The policy search space is flat: per (policy i, op j) one categorical op
index + two uniforms (prob, level) — see reference search.py:216-220.

Mechanics follow hyperopt.tpe:
  * first `n_startup` trials sample from the prior (uniform / flat choice)
  * afterwards observations are split by objective into good (best
    gamma-quantile) and bad; candidates are drawn from the good-model and
    ranked by log p_good(x) - log p_bad(x) (the EI surrogate)
  * uniform dims use an adaptive Parzen window: prior-augmented Gaussian
    mixture, sigma from neighbor spacing clipped to [range/min_sigma_div,
    range]; truncated to the bounds
  * choice dims use prior-smoothed categorical counts
Objective is MINIMIZED (callers pass -top1_valid).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Sequence, Tuple

import numpy as np


@dataclass
class ChoiceDim:
    name: str
    n: int


@dataclass
class UniformDim:
    name: str
    lo: float
    hi: float


class SearchSpace:
    def __init__(self, dims: Sequence):
        self.dims = list(dims)

    def sample_prior(self, rng: np.random.Generator) -> Dict:
        out = {}
        for d in self.dims:
            if isinstance(d, ChoiceDim):
                out[d.name] = int(rng.integers(0, d.n))
            else:
                out[d.name] = float(rng.uniform(d.lo, d.hi))
        return out


def _adaptive_parzen(mus: np.ndarray, lo: float, hi: float,
                     prior_weight: float = 1.0):
    """hyperopt's adaptive_parzen_normal: observed mus + a prior component
    at the range midpoint with sigma = range; per-component sigma from
    neighbor gaps, clipped."""
    prior_mu = 0.5 * (lo + hi)
    prior_sigma = hi - lo
    n = len(mus)
    if n == 0:
        return (np.array([prior_mu]), np.array([prior_sigma]), np.array([1.0]))
    order = np.argsort(mus)
    sorted_mus = mus[order]
    # neighbor-gap sigmas
    sigma = np.zeros(n)
    if n == 1:
        sigma[:] = prior_sigma
    else:
        left = np.concatenate([[sorted_mus[0] - lo], np.diff(sorted_mus)])
        right = np.concatenate([np.diff(sorted_mus), [hi - sorted_mus[-1]]])
        sigma = np.maximum(left, right)
    min_sigma = prior_sigma / min(100.0, 1.0 + n)
    sigma = np.clip(sigma, min_sigma, prior_sigma)
    # linear ramp weights down-weighting the OLDEST observations
    # (hyperopt linear_forgetting=25); mus arrive in time order, so build
    # time-ordered weights then permute into sorted order.
    lf = 25
    w_time = np.ones(n)
    if n > lf:
        w_time[: n - lf] = np.linspace(1.0 / n, 1.0, n - lf)
    w_sorted = w_time[order]
    mus_out = np.concatenate([[prior_mu], sorted_mus])
    sig_out = np.concatenate([[prior_sigma], sigma])
    w_out = np.concatenate([[prior_weight], w_sorted])
    w_out = w_out / w_out.sum()
    return mus_out, sig_out, w_out


def _gmm_logpdf(x: np.ndarray, mus: np.ndarray, sigmas: np.ndarray,
                ws: np.ndarray, lo: float, hi: float) -> np.ndarray:
    """log pdf of the truncated Gaussian mixture at points x."""
    from scipy.stats import norm
    x = np.atleast_1d(x)[:, None]
    mus, sigmas, ws = mus[None, :], sigmas[None, :], ws[None, :]
    # truncation normalizer per component
    z = norm.cdf((hi - mus) / sigmas) - norm.cdf((lo - mus) / sigmas)
    z = np.maximum(z, 1e-12)
    comp = ws * np.exp(-0.5 * ((x - mus) / sigmas) ** 2) / (sigmas * math.sqrt(2 * math.pi)) / z
    return np.log(np.maximum(comp.sum(axis=1), 1e-300))


def _gmm_sample(rng, n, mus, sigmas, ws, lo, hi) -> np.ndarray:
    idx = rng.choice(len(mus), size=n, p=ws)
    out = rng.normal(mus[idx], sigmas[idx])
    # resample out-of-bounds draws (cheap rejection; bounds are wide)
    for _ in range(50):
        bad = (out < lo) | (out > hi)
        if not bad.any():
            break
        out[bad] = rng.normal(mus[idx[bad]], sigmas[idx[bad]])
    return np.clip(out, lo, hi)


class TPESampler:
    def __init__(self, space: SearchSpace, seed: int = 0, gamma: float = 0.25,
                 n_startup: int = 20, n_candidates: int = 24,
                 prior_weight: float = 1.0):
        self.space = space
        self.rng = np.random.default_rng(seed)
        self.gamma = gamma
        self.n_startup = n_startup
        self.n_candidates = n_candidates
        self.prior_weight = prior_weight
        self.history: List[Tuple[Dict, float]] = []

    def observe(self, config: Dict, loss: float) -> None:
        self.history.append((config, float(loss)))

    def _split(self):
        losses = np.array([l for _, l in self.history])
        order = np.argsort(losses, kind="stable")
        # hyperopt: n_below = min(ceil(gamma*sqrt(n)), linear_forgetting)
        n_good = max(1, min(int(math.ceil(self.gamma * math.sqrt(len(losses)))), 25))
        good = [self.history[i][0] for i in order[:n_good]]
        bad = [self.history[i][0] for i in order[n_good:]]
        return good, bad

    def suggest(self) -> Dict:
        if len(self.history) < self.n_startup:
            return self.space.sample_prior(self.rng)
        good, bad = self._split()
        out = {}
        for d in self.space.dims:
            gv = np.array([g[d.name] for g in good])
            bv = np.array([b[d.name] for b in bad]) if bad else np.array([])
            if isinstance(d, ChoiceDim):
                out[d.name] = self._suggest_choice(d, gv.astype(int), bv.astype(int))
            else:
                out[d.name] = self._suggest_uniform(d, gv.astype(float), bv.astype(float))
        return out

    def _suggest_choice(self, d: ChoiceDim, good: np.ndarray, bad: np.ndarray) -> int:
        pg = np.bincount(good, minlength=d.n) + self.prior_weight
        pg = pg / pg.sum()
        pb = np.bincount(bad, minlength=d.n) + self.prior_weight
        pb = pb / pb.sum()
        cand = self.rng.choice(d.n, size=self.n_candidates, p=pg)
        score = np.log(pg[cand]) - np.log(pb[cand])
        return int(cand[np.argmax(score)])

    def _suggest_uniform(self, d: UniformDim, good: np.ndarray, bad: np.ndarray) -> float:
        mus_g, sig_g, w_g = _adaptive_parzen(good, d.lo, d.hi, self.prior_weight)
        mus_b, sig_b, w_b = _adaptive_parzen(bad, d.lo, d.hi, self.prior_weight)
        cand = _gmm_sample(self.rng, self.n_candidates, mus_g, sig_g, w_g, d.lo, d.hi)
        score = (_gmm_logpdf(cand, mus_g, sig_g, w_g, d.lo, d.hi)
                 - _gmm_logpdf(cand, mus_b, sig_b, w_b, d.lo, d.hi))
        return float(cand[np.argmax(score)])


def policy_search_space(num_policy: int, num_op: int, n_ops: int) -> SearchSpace:
    """The reference's space (search.py:216-220)."""
    dims = []
    for i in range(num_policy):
        for j in range(num_op):
            dims.append(ChoiceDim(f"policy_{i}_{j}", n_ops))
            dims.append(UniformDim(f"prob_{i}_{j}", 0.0, 1.0))
            dims.append(UniformDim(f"level_{i}_{j}", 0.0, 1.0))
    return SearchSpace(dims)
