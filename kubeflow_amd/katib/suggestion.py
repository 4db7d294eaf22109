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


def make_suggestion(algorithm: str, parameters: List[dict], seed: int = 0):
    alg = (algorithm or "random").lower()
    if alg in ("random",):
        return RandomSuggestion(parameters, seed)
    if alg in ("grid",):
        return GridSuggestion(parameters, seed)
    if alg in ("bayesianoptimization", "bayesopt", "skopt"):
        return BayesOptSuggestion(parameters, seed)
    raise ValueError(f"unknown suggestion algorithm {algorithm!r}")
