"""Tree-structured Parzen Estimator (TPE) hyperparameter sampler.

Reproduces the reference's search procedure — ``hyperopt.fmin(tpe.suggest,
max_evals=10)`` over {n_estimators ∈ choice(range(100,1000)), max_depth ∈
choice(range(1,25)), criterion ∈ {gini, entropy}} (01-train-model.ipynb
cell-8) — without hyperopt (unavailable offline). This is the actual TPE
algorithm (Bergstra et al., NeurIPS 2011), not a perturbation heuristic:

1. first ``n_startup`` trials are drawn uniformly from the space;
2. afterwards, observed trials are split by loss into the best γ-fraction
   ("good", density l(x)) and the rest ("bad", density g(x));
3. each dimension gets a 1-D Parzen mixture per split — Gaussian kernels
   with adaptive bandwidth plus a uniform prior component for integer
   dimensions, Dirichlet-smoothed category frequencies for categorical
   dimensions;
4. ``n_candidates`` draws from l(x) are scored by log l(x) − log g(x)
   (the expected-improvement surrogate) and the argmax is suggested.

One deliberate upgrade over the reference: hyperopt models the two integer
ranges as 900-way/24-way *categorical* choices, which makes its TPE
degenerate to near-random there; this sampler treats them as quantized
numeric dimensions so nearby values share density mass.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import numpy as np

__all__ = ["TPESampler", "IntDim", "CatDim"]


@dataclass(frozen=True)
class IntDim:
    """Integer dimension sampled on [lo, hi] inclusive."""

    lo: int
    hi: int


@dataclass(frozen=True)
class CatDim:
    """Categorical dimension over a fixed option tuple."""

    options: tuple


@dataclass
class _Trial:
    params: dict
    loss: float = field(default=np.inf)


class TPESampler:
    def __init__(
        self,
        space: dict,
        seed: int = 0,
        n_startup: int = 5,
        gamma: float = 0.25,
        n_candidates: int = 24,
    ):
        self.space = dict(space)
        self.rng = np.random.default_rng(seed)
        self.n_startup = int(n_startup)
        self.gamma = float(gamma)
        self.n_candidates = int(n_candidates)
        self.trials: list[_Trial] = []

    # ------------------------------------------------------------- sampling
    def suggest(self) -> dict:
        done = [t for t in self.trials if np.isfinite(t.loss)]
        if len(done) < self.n_startup:
            return self._draw_uniform()
        good, bad = self._split(done)
        cands = [self._draw_from_good(good) for _ in range(self.n_candidates)]
        scores = [self._ei_score(c, good, bad) for c in cands]
        return cands[int(np.argmax(scores))]

    def observe(self, params: dict, loss: float) -> None:
        """Record a completed trial (loss: lower is better, e.g. −roc_auc)."""
        self.trials.append(_Trial(dict(params), float(loss)))

    # ------------------------------------------------------------- internals
    def _draw_uniform(self) -> dict:
        out = {}
        for name, dim in self.space.items():
            if isinstance(dim, IntDim):
                out[name] = int(self.rng.integers(dim.lo, dim.hi + 1))
            else:
                out[name] = dim.options[int(self.rng.integers(len(dim.options)))]
        return out

    def _split(self, done: list[_Trial]) -> tuple[list[dict], list[dict]]:
        order = sorted(done, key=lambda t: t.loss)
        # hyperopt-style: at least 1, at most ceil(gamma * n) good trials
        n_good = max(1, int(np.ceil(self.gamma * len(order))))
        return (
            [t.params for t in order[:n_good]],
            [t.params for t in order[n_good:]] or [t.params for t in order],
        )

    # --- integer dims: adaptive-bandwidth Gaussian Parzen + uniform prior
    def _int_kernel(self, dim: IntDim, obs: list[int]):
        xs = np.asarray(sorted(obs), dtype=np.float64)
        span = float(dim.hi - dim.lo) or 1.0
        if len(xs) == 1:
            sig = np.array([span / 2.0])
        else:
            # bandwidth = max gap to the neighbours (hyperopt's heuristic),
            # clipped so kernels neither collapse nor flatten out
            gaps = np.empty_like(xs)
            gaps[0] = xs[1] - xs[0]
            gaps[-1] = xs[-1] - xs[-2]
            if len(xs) > 2:
                gaps[1:-1] = np.maximum(xs[1:-1] - xs[:-2], xs[2:] - xs[1:-1])
            sig = np.clip(gaps, span / min(100.0, 1.0 + len(xs) * 2), span)
        return xs, sig, span

    def _int_sample(self, dim: IntDim, obs: list[int]) -> int:
        xs, sig, _ = self._int_kernel(dim, obs)
        # uniform prior component gets one slot (weight 1/(n+1))
        k = int(self.rng.integers(len(xs) + 1))
        if k == len(xs):
            return int(self.rng.integers(dim.lo, dim.hi + 1))
        v = self.rng.normal(xs[k], sig[k])
        return int(np.clip(np.rint(v), dim.lo, dim.hi))

    def _int_logpdf(self, dim: IntDim, obs: list[int], x: int) -> float:
        xs, sig, span = self._int_kernel(dim, obs)
        z = (float(x) - xs) / sig
        comps = np.exp(-0.5 * z * z) / (sig * np.sqrt(2 * np.pi))
        dens = (comps.sum() + 1.0 / span) / (len(xs) + 1)
        return float(np.log(max(dens, 1e-300)))

    # --- categorical dims: Dirichlet-smoothed frequencies
    def _cat_probs(self, dim: CatDim, obs: list) -> np.ndarray:
        counts = np.ones(len(dim.options))  # +1 smoothing (uniform prior)
        index = {o: i for i, o in enumerate(dim.options)}
        for o in obs:
            counts[index[o]] += 1
        return counts / counts.sum()

    def _draw_from_good(self, good: list[dict]) -> dict:
        out = {}
        for name, dim in self.space.items():
            obs = [p[name] for p in good]
            if isinstance(dim, IntDim):
                out[name] = self._int_sample(dim, obs)
            else:
                probs = self._cat_probs(dim, obs)
                out[name] = dim.options[int(self.rng.choice(len(probs), p=probs))]
        return out

    def _ei_score(self, cand: dict, good: list[dict], bad: list[dict]) -> float:
        s = 0.0
        for name, dim in self.space.items():
            x = cand[name]
            og = [p[name] for p in good]
            ob = [p[name] for p in bad]
            if isinstance(dim, IntDim):
                s += self._int_logpdf(dim, og, x) - self._int_logpdf(dim, ob, x)
            else:
                i = dim.options.index(x)
                s += float(
                    np.log(self._cat_probs(dim, og)[i])
                    - np.log(self._cat_probs(dim, ob)[i])
                )
        return s


def reference_space() -> dict:
    """The reference search space (01-train cell-8)."""
    return {
        "n_estimators": IntDim(100, 999),
        "max_depth": IntDim(1, 24),
        "criterion": CatDim(("gini", "entropy")),
    }
