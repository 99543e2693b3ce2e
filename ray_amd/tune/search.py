"""Search algorithms (reference: python/ray/tune/search/ — Searcher
base, BasicVariantGenerator, ConcurrencyLimiter, and the OptunaSearch /
HyperOptSearch / BayesOptSearch adapters).

BayesOptSearch here is a NATIVE GP-based implementation (scikit-learn
is in the image); Optuna/HyperOpt bind lazily to their libraries and
raise a clear ImportError when absent.
"""
from __future__ import annotations

import math
import random
from typing import Any, Dict, List, Optional

from .impl import (
    Domain,
    _Choice,
    _GridSearch,
    _LogUniform,
    _QRandInt,
    _RandInt,
    _Randn,
    _SampleFrom,
    _Uniform,
    generate_variants,
)


class Searcher:
    """reference: tune/search/searcher.py — suggest()/on_trial_complete
    drive sequential config proposal with result feedback."""

    def __init__(self, metric: Optional[str] = None, mode: str = "max"):
        self.metric = metric
        self.mode = mode

    def set_search_properties(self, metric, mode, config) -> bool:
        if metric:
            self.metric = metric
        if mode:
            self.mode = mode
        return True

    def suggest(self, trial_id: str) -> Optional[Dict[str, Any]]:
        raise NotImplementedError

    def on_trial_complete(self, trial_id: str,
                          result: Optional[dict] = None,
                          error: bool = False) -> None:
        pass

    def on_trial_result(self, trial_id: str, result: dict) -> None:
        pass


class BasicVariantGenerator(Searcher):
    """Random/grid sampling — the reference's default search
    (search/basic_variant.py) expressed as a Searcher."""

    def __init__(self, metric=None, mode="max", seed: int = 0):
        super().__init__(metric, mode)
        self._space: Dict[str, Any] = {}
        self._drawn: List[dict] = []
        self._seed = seed

    def set_search_properties(self, metric, mode, config) -> bool:
        super().set_search_properties(metric, mode, config)
        if config:
            self._space = dict(config)
        return True

    def suggest(self, trial_id):
        # generate_variants is deterministic per seed, so variant idx
        # is stable across calls
        idx = len(self._drawn)
        cfg = generate_variants(self._space, idx + 1, seed=self._seed)[-1]
        self._drawn.append(cfg)
        return cfg


class ConcurrencyLimiter(Searcher):
    """Caps in-flight suggests (reference: search/concurrency_limiter
    .py) — suggest() returns None while `max_concurrent` trials are
    outstanding."""

    def __init__(self, searcher: Searcher, max_concurrent: int = 8):
        super().__init__(searcher.metric, searcher.mode)
        self.searcher = searcher
        self.max_concurrent = max_concurrent
        self._live: set = set()

    def set_search_properties(self, metric, mode, config) -> bool:
        return self.searcher.set_search_properties(metric, mode, config)

    def suggest(self, trial_id):
        if len(self._live) >= self.max_concurrent:
            return None
        cfg = self.searcher.suggest(trial_id)
        if cfg is not None:
            self._live.add(trial_id)
        return cfg

    def on_trial_complete(self, trial_id, result=None, error=False):
        self._live.discard(trial_id)
        self.searcher.on_trial_complete(trial_id, result, error)

    def on_trial_result(self, trial_id, result):
        self.searcher.on_trial_result(trial_id, result)


# ---------------------------------------------------------------------
# Native Bayesian optimization (GP + expected improvement)
# ---------------------------------------------------------------------


class BayesOptSearch(Searcher):
    """GP-EI over the numeric search space (reference:
    search/bayesopt/bayesopt_search.py, re-implemented natively on
    scikit-learn instead of the `bayesian-optimization` package).

    Numeric domains are mapped to [0,1] (log-scaled for loguniform);
    choices are one-hot-ish (index dimension). The first
    `n_startup_trials` configs are random, after which each suggest
    fits a GP on (x, metric) and maximizes expected improvement over
    random candidates.
    """

    def __init__(self, metric=None, mode="max", *,
                 n_startup_trials: int = 8, seed: int = 0,
                 n_candidates: int = 512):
        super().__init__(metric, mode)
        self._rng = random.Random(seed)
        self._np_seed = seed
        self.n_startup = n_startup_trials
        self.n_candidates = n_candidates
        self._space: Dict[str, Any] = {}
        self._dims: List[tuple] = []  # (key, kind, lo, hi / options)
        self._x: List[List[float]] = []
        self._y: List[float] = []
        self._pending: Dict[str, List[float]] = {}

    def set_search_properties(self, metric, mode, config) -> bool:
        super().set_search_properties(metric, mode, config)
        if not config:
            return True
        self._space = dict(config)
        self._dims = []
        for k, v in config.items():
            if isinstance(v, _Uniform):
                self._dims.append((k, "lin", float(v.lo), float(v.hi)))
            elif isinstance(v, _LogUniform):
                self._dims.append(
                    (k, "log", math.log(v.lo), math.log(v.hi)))
            elif isinstance(v, (_RandInt, _QRandInt)):
                self._dims.append((k, "int", float(v.lo), float(v.hi)))
            elif isinstance(v, _Randn):
                self._dims.append(
                    (k, "lin", v.mean - 4 * v.sd, v.mean + 4 * v.sd))
            elif isinstance(v, _Choice):
                self._dims.append((k, "choice", v.options, None))
            elif isinstance(v, (_GridSearch, _SampleFrom, Domain)):
                raise ValueError(
                    f"BayesOptSearch cannot optimize domain for {k!r}")
            # constants pass through
        return True

    # unit-cube <-> config
    def _decode(self, u: List[float]) -> Dict[str, Any]:
        cfg = {k: v for k, v in self._space.items()
               if not isinstance(v, (Domain, _GridSearch))}
        for (k, kind, a, b), t in zip(self._dims, u):
            if kind == "choice":
                cfg[k] = a[min(int(t * len(a)), len(a) - 1)]
            elif kind == "int":
                cfg[k] = int(round(a + t * (b - a)))
            elif kind == "log":
                cfg[k] = math.exp(a + t * (b - a))
            else:
                cfg[k] = a + t * (b - a)
        return cfg

    def _random_u(self) -> List[float]:
        return [self._rng.random() for _ in self._dims]

    def suggest(self, trial_id):
        if not self._dims:
            raise RuntimeError("search space not set (pass param_space)")
        if len(self._x) < self.n_startup:
            u = self._random_u()
        else:
            u = self._ei_argmax()
        self._pending[trial_id] = u
        return self._decode(u)

    def _ei_argmax(self) -> List[float]:
        import numpy as np
        from sklearn.gaussian_process import GaussianProcessRegressor
        from sklearn.gaussian_process.kernels import RBF, ConstantKernel

        X = np.asarray(self._x)
        y = np.asarray(self._y, dtype=float)
        if self.mode == "min":
            y = -y
        yn = (y - y.mean()) / (y.std() + 1e-9)
        gp = GaussianProcessRegressor(
            kernel=ConstantKernel(1.0) * RBF(0.25),
            alpha=1e-4, normalize_y=False,
            random_state=self._np_seed)
        gp.fit(X, yn)
        rng = np.random.RandomState(self._np_seed + len(self._x))
        cand = rng.rand(self.n_candidates, len(self._dims))
        mu, sd = gp.predict(cand, return_std=True)
        best = yn.max()
        sd = np.maximum(sd, 1e-9)
        z = (mu - best) / sd
        from scipy.stats import norm

        ei = (mu - best) * norm.cdf(z) + sd * norm.pdf(z)
        return [float(v) for v in cand[int(np.argmax(ei))]]

    def on_trial_complete(self, trial_id, result=None, error=False):
        u = self._pending.pop(trial_id, None)
        if u is None or error or not result:
            return
        val = result.get(self.metric) if self.metric else None
        if val is None:
            return
        self._x.append(u)
        self._y.append(float(val))


# ---------------------------------------------------------------------
# Optuna / HyperOpt adapters (lazy imports — optional libraries)
# ---------------------------------------------------------------------


class OptunaSearch(Searcher):
    """reference: search/optuna/optuna_search.py — ask/tell against an
    optuna.Study built from the Tune search space."""

    def __init__(self, metric=None, mode="max", *, seed: Optional[int] = None,
                 sampler=None):
        super().__init__(metric, mode)
        try:
            import optuna
        except ImportError as e:
            raise ImportError(
                "OptunaSearch requires the `optuna` package") from e
        self._optuna = optuna
        self._sampler = sampler or optuna.samplers.TPESampler(seed=seed)
        self._study = None
        self._space: Dict[str, Any] = {}
        self._trials: Dict[str, Any] = {}

    def set_search_properties(self, metric, mode, config) -> bool:
        super().set_search_properties(metric, mode, config)
        if config:
            self._space = dict(config)
        direction = "minimize" if self.mode == "min" else "maximize"
        self._study = self._optuna.create_study(
            sampler=self._sampler, direction=direction)
        return True

    def _ask_dim(self, trial, k, v):
        if isinstance(v, _Uniform):
            return trial.suggest_float(k, v.lo, v.hi)
        if isinstance(v, _LogUniform):
            return trial.suggest_float(k, v.lo, v.hi, log=True)
        if isinstance(v, (_RandInt, _QRandInt)):
            return trial.suggest_int(k, v.lo, v.hi - 1)
        if isinstance(v, _Choice):
            return trial.suggest_categorical(k, v.options)
        if isinstance(v, _Randn):
            return trial.suggest_float(k, v.mean - 4 * v.sd,
                                       v.mean + 4 * v.sd)
        return v

    def suggest(self, trial_id):
        if self._study is None:
            self.set_search_properties(self.metric, self.mode, self._space)
        t = self._study.ask()
        self._trials[trial_id] = t
        cfg = {}
        for k, v in self._space.items():
            cfg[k] = self._ask_dim(t, k, v) if isinstance(v, Domain) else v
        return cfg

    def on_trial_complete(self, trial_id, result=None, error=False):
        t = self._trials.pop(trial_id, None)
        if t is None:
            return
        if error or not result or (self.metric and
                                   result.get(self.metric) is None):
            self._study.tell(t, state=self._optuna.trial.TrialState.FAIL)
            return
        self._study.tell(t, float(result[self.metric]))


class HyperOptSearch(Searcher):
    """reference: search/hyperopt/hyperopt_search.py — TPE via
    hyperopt.fmin's ask/tell internals (Trials object)."""

    def __init__(self, metric=None, mode="max", *,
                 n_initial_points: int = 20, seed: Optional[int] = None):
        super().__init__(metric, mode)
        try:
            import hyperopt
        except ImportError as e:
            raise ImportError(
                "HyperOptSearch requires the `hyperopt` package") from e
        self._hpo = hyperopt
        self._n_init = n_initial_points
        self._seed = seed
        self._space: Dict[str, Any] = {}
        self._hp_space = None
        self._trials = None
        self._ids: Dict[str, int] = {}

    def set_search_properties(self, metric, mode, config) -> bool:
        super().set_search_properties(metric, mode, config)
        hp = self._hpo.hp
        if config:
            self._space = dict(config)
        hspace = {}
        for k, v in self._space.items():
            if isinstance(v, _Uniform):
                hspace[k] = hp.uniform(k, v.lo, v.hi)
            elif isinstance(v, _LogUniform):
                hspace[k] = hp.loguniform(k, math.log(v.lo), math.log(v.hi))
            elif isinstance(v, (_RandInt, _QRandInt)):
                hspace[k] = hp.randint(k, v.lo, v.hi)
            elif isinstance(v, _Choice):
                hspace[k] = hp.choice(k, v.options)
            elif isinstance(v, _Randn):
                hspace[k] = hp.normal(k, v.mean, v.sd)
        self._hp_space = hspace
        self._trials = self._hpo.Trials()
        return True

    def suggest(self, trial_id):
        if self._trials is None:
            self.set_search_properties(self.metric, self.mode, self._space)
        hpo = self._hpo
        n = len(self._trials.trials)
        algo = (hpo.rand.suggest if n < self._n_init else hpo.tpe.suggest)
        new = algo(
            [n], hpo.base.Domain(lambda spc: 0, self._hp_space),
            self._trials,
            self._seed if self._seed is not None else n,
        )
        self._trials.insert_trial_docs(new)
        self._trials.refresh()
        self._ids[trial_id] = n
        vals = {k: v[0] for k, v in new[0]["misc"]["vals"].items() if v}
        cfg = {k: v for k, v in self._space.items()
               if not isinstance(v, Domain)}
        for k, dv in self._space.items():
            if not isinstance(dv, Domain):
                continue
            raw = vals.get(k)
            if isinstance(dv, _Choice):
                cfg[k] = dv.options[int(raw)]
            elif isinstance(dv, (_RandInt, _QRandInt)):
                cfg[k] = int(raw)
            else:
                cfg[k] = float(raw)
        return cfg

    def on_trial_complete(self, trial_id, result=None, error=False):
        idx = self._ids.pop(trial_id, None)
        if idx is None or idx >= len(self._trials.trials):
            return
        t = self._trials.trials[idx]
        if error or not result or (self.metric and
                                   result.get(self.metric) is None):
            t["state"] = self._hpo.JOB_STATE_ERROR
        else:
            val = float(result[self.metric])
            loss = -val if self.mode == "max" else val
            t["result"] = {"loss": loss, "status": self._hpo.STATUS_OK}
            t["state"] = self._hpo.JOB_STATE_DONE
        self._trials.refresh()
