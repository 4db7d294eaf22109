"""Katib suggestion service — hyperparameter proposal algorithms.

The reference drives Katib StudyJobs via the custom-objects API
(testing/katib_studyjob_test.py:39-120); the suggestion service lives in the
sibling katib repo. Here it is in-process: given the experiment's parameter
space and observed trials, propose the next assignments.

Algorithms: random, grid, and bayesian-lite (GP with RBF kernel + expected
improvement via scikit-learn, available offline).

Parameter space shape (Katib v1beta1 parameters):
    [{"name": "lr", "parameterType": "double",
      "feasibleSpace": {"min": "1e-4", "max": "1e-1", "logScale": true}},
     {"name": "hidden", "parameterType": "int",
      "feasibleSpace": {"min": "64", "max": "512"}},
     {"name": "opt", "parameterType": "categorical",
      "feasibleSpace": {"list": ["adamw", "sgd"]}}]
"""
from __future__ import annotations

import itertools
import math
import random
from typing import Dict, List, Optional, Tuple


class _Space:
    def __init__(self, parameters: List[dict]):
        self.params = parameters

    def sample(self, rng: random.Random) -> Dict[str, object]:
        out = {}
        for p in self.params:
            fs = p.get("feasibleSpace", {})
            t = p.get("parameterType", "double")
            if t == "categorical":
                out[p["name"]] = rng.choice(fs["list"])
            elif t == "int":
                out[p["name"]] = rng.randint(int(fs["min"]), int(fs["max"]))
            else:
                lo, hi = float(fs["min"]), float(fs["max"])
                if fs.get("logScale"):
                    out[p["name"]] = math.exp(
                        rng.uniform(math.log(lo), math.log(hi)))
                else:
                    out[p["name"]] = rng.uniform(lo, hi)
        return out

    def to_unit(self, assignment: Dict[str, object]) -> List[float]:
        """Encode an assignment into [0,1]^d for the GP."""
        vec = []
        for p in self.params:
            fs = p.get("feasibleSpace", {})
            t = p.get("parameterType", "double")
            v = assignment[p["name"]]
            if t == "categorical":
                vec.append(fs["list"].index(v) / max(1, len(fs["list"]) - 1))
            else:
                lo, hi = float(fs["min"]), float(fs["max"])
                if fs.get("logScale"):
                    vec.append((math.log(float(v)) - math.log(lo)) /
                               max(1e-12, math.log(hi) - math.log(lo)))
                else:
                    vec.append((float(v) - lo) / max(1e-12, hi - lo))
        return vec


    def from_unit(self, vec: List[float]) -> Dict[str, object]:
        """Decode a [0,1]^d point back into an assignment (inverse of
        to_unit; ints rounded, categoricals to the nearest index)."""
        out = {}
        for u, p in zip(vec, self.params):
            fs = p.get("feasibleSpace", {})
            t = p.get("parameterType", "double")
            u = min(1.0, max(0.0, float(u)))
            if t == "categorical":
                lst = fs["list"]
                out[p["name"]] = lst[round(u * (len(lst) - 1))]
                continue
            lo, hi = float(fs["min"]), float(fs["max"])
            if fs.get("logScale"):
                v = math.exp(math.log(lo) + u * (math.log(hi) - math.log(lo)))
            else:
                v = lo + u * (hi - lo)
            v = min(hi, max(lo, v))  # exp/log round-trip can overshoot 1 ulp
            out[p["name"]] = (min(int(fs["max"]), max(int(fs["min"]),
                                                      round(v)))
                              if t == "int" else v)
        return out


class RandomSuggestion:
    def __init__(self, parameters: List[dict], seed: int = 0):
        self.space = _Space(parameters)
        self.rng = random.Random(seed)

    def suggest(self, trials: List[Tuple[dict, Optional[float]]],
                n: int) -> List[dict]:
        return [self.space.sample(self.rng) for _ in range(n)]


class GridSuggestion:
    def __init__(self, parameters: List[dict], seed: int = 0,
                 points_per_dim: int = 4):
        self.space = _Space(parameters)
        axes = []
        for p in parameters:
            fs = p.get("feasibleSpace", {})
            t = p.get("parameterType", "double")
            if t == "categorical":
                axes.append(list(fs["list"]))
            elif t == "int":
                lo, hi = int(fs["min"]), int(fs["max"])
                step = max(1, (hi - lo) // (points_per_dim - 1 or 1))
                axes.append(list(range(lo, hi + 1, step)))
            else:
                lo, hi = float(fs["min"]), float(fs["max"])
                if fs.get("logScale"):
                    axes.append([math.exp(math.log(lo) + i *
                                          (math.log(hi) - math.log(lo)) /
                                          (points_per_dim - 1))
                                 for i in range(points_per_dim)])
                else:
                    axes.append([lo + i * (hi - lo) / (points_per_dim - 1)
                                 for i in range(points_per_dim)])
        self.grid = [dict(zip([p["name"] for p in parameters], combo))
                     for combo in itertools.product(*axes)]
        self.cursor = 0

    def suggest(self, trials, n):
        out = []
        while len(out) < n and self.cursor < len(self.grid):
            out.append(self.grid[self.cursor])
            self.cursor += 1
        return out


class BayesOptSuggestion:
    """GP + expected-improvement over random candidates (minimization)."""

    def __init__(self, parameters: List[dict], seed: int = 0,
                 n_initial: int = 4, n_candidates: int = 256):
        self.space = _Space(parameters)
        self.rng = random.Random(seed)
        self.n_initial = n_initial
        self.n_candidates = n_candidates

    def suggest(self, trials, n):
        done = [(a, v) for a, v in trials if v is not None]
        if len(done) < self.n_initial:
            return [self.space.sample(self.rng) for _ in range(n)]
        try:
            import numpy as np
            from sklearn.gaussian_process import GaussianProcessRegressor
            from sklearn.gaussian_process.kernels import Matern
            from scipy.stats import norm
        except ImportError:  # pragma: no cover
            return [self.space.sample(self.rng) for _ in range(n)]
        X = np.array([self.space.to_unit(a) for a, _ in done])
        y = np.array([v for _, v in done], dtype=float)
        gp = GaussianProcessRegressor(kernel=Matern(nu=2.5),
                                      normalize_y=True, alpha=1e-6)
        gp.fit(X, y)
        best = y.min()
        out = []
        for _ in range(n):
            cands = [self.space.sample(self.rng)
                     for _ in range(self.n_candidates)]
            Xc = np.array([self.space.to_unit(c) for c in cands])
            mu, sigma = gp.predict(Xc, return_std=True)
            sigma = np.maximum(sigma, 1e-9)
            imp = best - mu
            z = imp / sigma
            ei = imp * norm.cdf(z) + sigma * norm.pdf(z)
            out.append(cands[int(ei.argmax())])
        return out


class TpeSuggestion:
    """Tree-structured Parzen estimator (minimization). Completed trials
    split at the gamma quantile into good/bad sets; each is modeled as a
    Gaussian mixture (kernel centered on every observation, fixed unit-cube
    bandwidth); candidates sampled around the good set are ranked by
    log l(x) - log g(x). Mirrors Katib's `tpe` algorithm (hyperopt-style)
    without the hyperopt dependency."""

    def __init__(self, parameters: List[dict], seed: int = 0,
                 gamma: float = 0.25, n_initial: int = 4,
                 n_candidates: int = 128, bandwidth: float = 0.15):
        self.space = _Space(parameters)
        self.rng = random.Random(seed)
        self.seed = seed
        self.gamma = gamma
        self.n_initial = n_initial
        self.n_candidates = n_candidates
        self.bw = bandwidth

    def suggest(self, trials, n):
        done = [(a, v) for a, v in trials if v is not None]
        if len(done) < max(self.n_initial, 2):
            return [self.space.sample(self.rng) for _ in range(n)]
        import numpy as np
        npr = np.random.default_rng(self.seed + len(done))
        X = np.array([self.space.to_unit(a) for a, _ in done])
        y = np.array([v for _, v in done], dtype=float)
        order = np.argsort(y)
        n_good = max(1, int(math.ceil(self.gamma * len(done))))
        good, bad = X[order[:n_good]], X[order[n_good:]]
        if not len(bad):
            return [self.space.sample(self.rng) for _ in range(n)]
        d = X.shape[1]

        def log_mix(pts, data):
            diff = (pts[:, None, :] - data[None, :, :]) / self.bw
            ll = (-0.5 * (diff ** 2).sum(-1)
                  - d * math.log(self.bw * math.sqrt(2 * math.pi)))
            m = ll.max(axis=1)
            return m + np.log(np.exp(ll - m[:, None]).mean(axis=1))

        out = []
        for _ in range(n):
            centers = good[npr.integers(0, len(good), self.n_candidates)]
            cand = np.clip(centers + npr.normal(0, self.bw, centers.shape),
                           0.0, 1.0)
            cand[:self.n_candidates // 4] = npr.random(
                (self.n_candidates // 4, d))  # keep exploring
            score = log_mix(cand, good) - log_mix(cand, bad)
            out.append(self.space.from_unit(cand[int(score.argmax())]))
        return out


def make_suggestion(algorithm: str, parameters: List[dict], seed: int = 0):
    alg = (algorithm or "random").lower()
    if alg in ("random",):
        return RandomSuggestion(parameters, seed)
    if alg in ("grid",):
        return GridSuggestion(parameters, seed)
    if alg in ("bayesianoptimization", "bayesopt", "skopt"):
        return BayesOptSuggestion(parameters, seed)
    if alg in ("tpe", "hyperopt-tpe"):
        return TpeSuggestion(parameters, seed)
    raise ValueError(f"unknown suggestion algorithm {algorithm!r}")
