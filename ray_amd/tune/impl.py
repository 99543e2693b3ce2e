"""Tune implementation.

Reference counterparts: tune/tuner.py:43 Tuner (.fit :332),
tune/execution/tune_controller.py (actor-based trial loop),
tune/search/basic_variant.py (grid/random variant generation),
tune/schedulers/async_hyperband.py ASHA, tune/result_grid.py.
"""
from __future__ import annotations

import itertools
import math
import os
import random
import threading
import time
import traceback
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

import numpy as np

from ..train.checkpoint import Checkpoint
from ..train.config import Result, RunConfig
from ..train.session import TrainSession, _set_session

# ---------------- search space primitives ----------------


class Domain:
    def sample(self, rng: random.Random):
        raise NotImplementedError


class _Uniform(Domain):
    def __init__(self, lo, hi):
        self.lo, self.hi = lo, hi

    def sample(self, rng):
        return rng.uniform(self.lo, self.hi)


class _LogUniform(Domain):
    def __init__(self, lo, hi):
        self.lo, self.hi = lo, hi

    def sample(self, rng):
        return math.exp(rng.uniform(math.log(self.lo), math.log(self.hi)))


class _Randn(Domain):
    def __init__(self, mean, sd):
        self.mean, self.sd = mean, sd

    def sample(self, rng):
        return rng.gauss(self.mean, self.sd)


class _RandInt(Domain):
    def __init__(self, lo, hi):
        self.lo, self.hi = lo, hi

    def sample(self, rng):
        return rng.randrange(self.lo, self.hi)


class _QRandInt(Domain):
    def __init__(self, lo, hi, q):
        self.lo, self.hi, self.q = lo, hi, q

    def sample(self, rng):
        return (rng.randrange(self.lo, self.hi + 1) // self.q) * self.q


class _Choice(Domain):
    def __init__(self, options):
        self.options = list(options)

    def sample(self, rng):
        return rng.choice(self.options)


class _SampleFrom(Domain):
    def __init__(self, fn):
        self.fn = fn

    def sample(self, rng):
        return self.fn(None)


class _GridSearch:
    def __init__(self, values):
        self.values = list(values)


def uniform(lo, hi):
    return _Uniform(lo, hi)


def loguniform(lo, hi):
    return _LogUniform(lo, hi)


def randn(mean=0.0, sd=1.0):
    return _Randn(mean, sd)


def randint(lo, hi):
    return _RandInt(lo, hi)


def qrandint(lo, hi, q=1):
    return _QRandInt(lo, hi, q)


def choice(options):
    return _Choice(options)


def sample_from(fn):
    return _SampleFrom(fn)


def grid_search(values):
    return _GridSearch(values)


def generate_variants(space: Dict[str, Any], num_samples: int, seed: int = 0
                      ) -> List[Dict[str, Any]]:
    """Basic variant generator: cartesian product of grid_search values,
    each repeated num_samples times with random Domains resampled."""
    rng = random.Random(seed)
    grid_keys = [k for k, v in space.items() if isinstance(v, _GridSearch)]
    grids = [space[k].values for k in grid_keys]
    combos = list(itertools.product(*grids)) if grid_keys else [()]
    variants = []
    for _ in range(num_samples):
        for combo in combos:
            cfg = {}
            for k, v in space.items():
                if isinstance(v, _GridSearch):
                    cfg[k] = combo[grid_keys.index(k)]
                elif isinstance(v, Domain):
                    cfg[k] = v.sample(rng)
                elif isinstance(v, dict):
                    cfg[k] = generate_variants(v, 1, rng.randrange(1 << 30))[0]
                else:
                    cfg[k] = v
            variants.append(cfg)
    return variants


# ---------------- schedulers ----------------


class FIFOScheduler:
    def on_trial_result(self, trial_id: str, iteration: int, metric_value) -> str:
        return "CONTINUE"


class ASHAScheduler:
    """Async successive halving (reference:
    tune/schedulers/async_hyperband.py): rungs at grace_period *
    reduction_factor^k; a trial stops at a rung if it is below the top
    1/reduction_factor quantile of completed results at that rung."""

    def __init__(self, metric: Optional[str] = None, mode: str = "max",
                 max_t: int = 100, grace_period: int = 1,
                 reduction_factor: float = 3, time_attr="training_iteration"):
        self.metric = metric
        self.mode = mode
        self.max_t = max_t
        self.grace = grace_period
        self.rf = reduction_factor
        self.rungs: Dict[int, List[float]] = {}
        r = grace_period
        while r < max_t:
            self.rungs[int(r)] = []
            r *= reduction_factor

    def on_trial_result(self, trial_id, iteration, metric_value) -> str:
        if metric_value is None:
            return "CONTINUE"
        v = float(metric_value) if self.mode == "max" else -float(metric_value)
        if iteration >= self.max_t:
            return "STOP"
        rung = None
        for r in sorted(self.rungs, reverse=True):
            if iteration >= r:
                rung = r
                break
        if rung is None:
            return "CONTINUE"
        recorded = self.rungs[rung]
        recorded.append(v)
        k = max(1, int(len(recorded) / self.rf))
        cutoff = sorted(recorded, reverse=True)[k - 1]
        return "CONTINUE" if v >= cutoff else "STOP"


# ---------------- config ----------------


class HyperBandScheduler:
    """HyperBand (reference: tune/schedulers/hyperband.py): trials are
    assigned round-robin to brackets with geometrically increasing
    grace periods (r, r*eta, ...); within a bracket, successive
    halving keeps the top 1/eta at each rung. The bracket diversity is
    what distinguishes this from plain ASHA — aggressive brackets
    stop early, conservative ones let slow starters run."""

    def __init__(self, *, metric=None, mode: str = "max",
                 max_t: int = 81, reduction_factor: float = 3,
                 time_attr="training_iteration"):
        self.metric = metric
        self.mode = mode
        self.max_t = max_t
        self.eta = reduction_factor
        # bracket i has grace period eta^i (capped below max_t)
        self.n_brackets = max(1, int(math.log(max_t, reduction_factor)))
        self._bracket_of: Dict[str, int] = {}
        self._next_bracket = 0
        # (bracket, rung) -> recorded values
        self._rungs: Dict[tuple, List[float]] = {}

    def _bracket(self, trial_id: str) -> int:
        b = self._bracket_of.get(trial_id)
        if b is None:
            b = self._next_bracket % self.n_brackets
            self._next_bracket += 1
            self._bracket_of[trial_id] = b
        return b

    def on_trial_result(self, trial_id, iteration, metric_value) -> str:
        if metric_value is None:
            return "CONTINUE"
        if iteration >= self.max_t:
            return "STOP"
        v = float(metric_value) if self.mode == "max" else -float(
            metric_value)
        b = self._bracket(trial_id)
        grace = self.eta ** b
        rung = None
        r = grace
        while r <= iteration and r < self.max_t:
            rung = r
            r *= self.eta
        if rung is None or iteration != int(rung):
            return "CONTINUE"
        rec = self._rungs.setdefault((b, int(rung)), [])
        rec.append(v)
        k = max(1, int(len(rec) / self.eta))
        cutoff = sorted(rec, reverse=True)[k - 1]
        return "CONTINUE" if v >= cutoff else "STOP"


class MedianStoppingRule:
    """Stop trials whose running mean falls below the median of all
    trials' running means at the same iteration (reference:
    tune/schedulers/median_stopping_rule.py)."""

    def __init__(self, *, metric=None, mode: str = "max",
                 grace_period: int = 4, min_samples_required: int = 3):
        self.metric = metric
        self.mode = mode
        self.grace_period = grace_period
        self.min_samples = min_samples_required
        self._history: Dict[str, list] = {}

    def on_trial_result(self, trial_id, iteration, metric_value) -> str:
        if metric_value is None:
            return "CONTINUE"
        h = self._history.setdefault(trial_id, [])
        h.append(float(metric_value))
        if iteration < self.grace_period:
            return "CONTINUE"
        if len(self._history) < self.min_samples:
            return "CONTINUE"
        import statistics

        means = [
            sum(v) / len(v) for k, v in self._history.items() if v
        ]
        if len(means) < self.min_samples:
            return "CONTINUE"
        med = statistics.median(means)
        mine = sum(h) / len(h)
        worse = mine < med if self.mode == "max" else mine > med
        return "STOP" if worse else "CONTINUE"


class PopulationBasedTraining:
    """PBT (reference: tune/schedulers/pbt.py PopulationBasedTraining).

    Every `perturbation_interval` iterations a bottom-quantile trial is
    restarted from a top-quantile trial's latest checkpoint with a
    perturbed clone of its config (exploit + explore)."""

    def __init__(self, *, time_attr="training_iteration", metric=None,
                 mode="max", perturbation_interval=4,
                 hyperparam_mutations=None, quantile_fraction=0.25,
                 resample_probability=0.25, seed=0):
        self.metric = metric
        self.mode = mode
        self.perturbation_interval = perturbation_interval
        self.hyperparam_mutations = hyperparam_mutations or {}
        self.quantile_fraction = quantile_fraction
        self.resample_probability = resample_probability
        self._rng = random.Random(seed)
        self._scores: Dict[str, tuple] = {}       # trial -> (it, score)
        self._last_perturb: Dict[str, int] = {}
        self.num_perturbations = 0

    def on_trial_result(self, trial_id, iteration, metric_value) -> str:
        if metric_value is not None:
            self._scores[trial_id] = (iteration, metric_value)
        return "CONTINUE"

    def should_perturb(self, trial_id, iteration) -> bool:
        return (
            iteration - self._last_perturb.get(trial_id, 0)
            >= self.perturbation_interval
        )

    def exploit_target(self, trial_id) -> Optional[str]:
        """Donor trial if `trial_id` is in the bottom quantile, else None."""
        if len(self._scores) < 2:
            return None
        items = sorted(
            self._scores.items(), key=lambda kv: kv[1][1],
            reverse=(self.mode == "max"),
        )
        q = max(1, int(len(items) * self.quantile_fraction))
        top = [k for k, _ in items[:q]]
        bottom = {k for k, _ in items[-q:]}
        if trial_id not in bottom or trial_id in top:
            return None
        return self._rng.choice(top)

    def explore(self, config: dict) -> dict:
        new = dict(config)
        for k, dom in self.hyperparam_mutations.items():
            if self._rng.random() < self.resample_probability:
                if isinstance(dom, Domain):
                    new[k] = dom.sample(self._rng)
                elif isinstance(dom, list):
                    new[k] = self._rng.choice(dom)
                elif callable(dom):
                    new[k] = dom()
                continue
            v = new.get(k)
            if isinstance(dom, list) and v in dom:
                i = dom.index(v) + self._rng.choice([-1, 1])
                new[k] = dom[max(0, min(len(dom) - 1, i))]
            elif isinstance(v, bool):
                new[k] = not v if self._rng.random() < 0.5 else v
            elif isinstance(v, (int, float)):
                f = 1.2 if self._rng.random() < 0.5 else 0.8
                new[k] = type(v)(v * f) if isinstance(v, int) else v * f
        return new


class PB2(PopulationBasedTraining):
    """PB2 (reference: tune/schedulers/pb2.py): PBT where explore()
    picks new hyperparameters by GP-UCB over the observed
    (config -> reward-improvement) data instead of random
    perturbation — sample-efficient for small populations.
    hyperparam_bounds: {name: [low, high]} (continuous)."""

    def __init__(self, *, hyperparam_bounds=None, ucb_kappa=1.5, **kw):
        kw.setdefault("hyperparam_mutations", {})
        super().__init__(**kw)
        self.hyperparam_bounds = hyperparam_bounds or {}
        self.ucb_kappa = ucb_kappa
        self._history = []        # (xvec, reward_delta)
        self._prev_score: Dict[str, float] = {}
        self._trial_cfgs: Dict[str, dict] = {}

    def on_trial_result(self, trial_id, iteration, metric_value) -> str:
        if metric_value is not None:
            prev = self._prev_score.get(trial_id)
            if prev is not None:
                # record improvement for the trial's current config
                cfg = self._trial_cfgs.get(trial_id)
                if cfg is not None:
                    x = self._vec(cfg)
                    if x is not None:
                        delta = metric_value - prev
                        if self.mode == "min":
                            delta = -delta
                        self._history.append((x, delta))
            self._prev_score[trial_id] = metric_value
        return super().on_trial_result(trial_id, iteration, metric_value)

    def observe_config(self, trial_id, config):
        self._trial_cfgs[trial_id] = dict(config)

    def _vec(self, cfg):
        try:
            return [float(cfg[k]) for k in sorted(self.hyperparam_bounds)]
        except (KeyError, TypeError, ValueError):
            return None

    def explore(self, config: dict) -> dict:
        keys = sorted(self.hyperparam_bounds)
        if not keys or len(self._history) < 4:
            return super().explore(config)
        try:
            import numpy as _np
            from sklearn.gaussian_process import GaussianProcessRegressor
            from sklearn.gaussian_process.kernels import Matern
        except ImportError:
            return super().explore(config)
        X = _np.array([x for x, _ in self._history[-64:]])
        y = _np.array([d for _, d in self._history[-64:]])
        lo = _np.array([self.hyperparam_bounds[k][0] for k in keys])
        hi = _np.array([self.hyperparam_bounds[k][1] for k in keys])
        span = _np.where(hi > lo, hi - lo, 1.0)
        gp = GaussianProcessRegressor(
            kernel=Matern(nu=2.5), alpha=1e-4, normalize_y=True)
        gp.fit((X - lo) / span, y)
        cand = _np.random.default_rng(
            self._rng.randrange(1 << 30)).uniform(0, 1, size=(256, len(keys)))
        mu, sd = gp.predict(cand, return_std=True)
        best = cand[int(_np.argmax(mu + self.ucb_kappa * sd))]
        new = dict(config)
        for k, v01, l, s in zip(keys, best, lo, span):
            val = float(l + v01 * s)
            if isinstance(config.get(k), int):
                val = int(round(val))
            new[k] = val
        return new


@dataclass
class TuneConfig:
    metric: Optional[str] = None
    mode: str = "max"
    num_samples: int = 1
    max_concurrent_trials: Optional[int] = None
    scheduler: Any = None
    search_alg: Any = None
    time_budget_s: Optional[float] = None


# ---------------- trial runner actor ----------------


class _TrialActor:
    def __init__(self):
        self.session: Optional[TrainSession] = None
        self.done = False
        self.error: Optional[str] = None

    def run(self, fn_bytes: bytes, config: dict, storage_dir: str,
            trial_name: str, restore_path: Optional[str] = None):
        import cloudpickle

        fn = cloudpickle.loads(fn_bytes)
        os.makedirs(storage_dir, exist_ok=True)
        ckpt = Checkpoint(restore_path) if restore_path else None
        self.session = TrainSession(0, 1, 0, 1, storage_dir, trial_name,
                                    latest_checkpoint=ckpt)
        self.done = False

        def _run():
            _set_session(self.session)
            try:
                if isinstance(fn, type):
                    # class Trainable: step() loop until scheduler stops us
                    inst = fn(config)
                    while not self.session.stop_requested.is_set():
                        res = inst.step()
                        self.session.report(res)
                        if res.get("done"):
                            break
                else:
                    fn(config)
            except BaseException:
                self.error = traceback.format_exc()
            finally:
                self.done = True
                _set_session(None)

        threading.Thread(target=_run, daemon=True).start()
        return True

    def fetch(self):
        out = []
        if self.session:
            while not self.session.results_queue.empty():
                out.append(self.session.results_queue.get_nowait())
        return {"results": out, "done": self.done, "error": self.error}

    def request_stop(self):
        if self.session:
            self.session.stop_requested.set()
        return True

    def latest_checkpoint_path(self):
        if self.session and self.session.latest_checkpoint:
            return self.session.latest_checkpoint.path
        return None


# ---------------- Tuner ----------------


class ResultGrid:
    def __init__(self, results: List[Result], metric=None, mode="max"):
        self._results = results
        self._metric = metric
        self._mode = mode

    def __len__(self):
        return len(self._results)

    def __getitem__(self, i):
        return self._results[i]

    def __iter__(self):
        return iter(self._results)

    @property
    def errors(self):
        return [r.error for r in self._results if r.error]

    def get_best_result(self, metric: Optional[str] = None,
                        mode: Optional[str] = None) -> Result:
        metric = metric or self._metric
        mode = mode or self._mode
        valid = [r for r in self._results
                 if r.metrics and metric in r.metrics]
        if not valid:
            raise ValueError(f"no results with metric {metric!r}")
        key = lambda r: r.metrics[metric]  # noqa: E731
        return max(valid, key=key) if mode == "max" else min(valid, key=key)

    def get_dataframe(self):
        import pandas as pd

        return pd.DataFrame([r.metrics for r in self._results if r.metrics])


class Tuner:
    def __init__(self, trainable, *, param_space: Optional[dict] = None,
                 tune_config: Optional[TuneConfig] = None,
                 run_config: Optional[RunConfig] = None):
        self._trainable = trainable
        self.param_space = param_space or {}
        self.tune_config = tune_config or TuneConfig()
        self.run_config = run_config or RunConfig(name=f"tune_{int(time.time())}")
        self._restored: Optional[dict] = None  # Tuner.restore state

    @classmethod
    def restore(cls, path: str, trainable=None) -> "Tuner":
        """Resume an interrupted experiment (reference: Tuner.restore):
        finished trials load from disk, unfinished ones re-run."""
        import cloudpickle

        with open(os.path.join(path, "tuner_state.pkl"), "rb") as f:
            state = cloudpickle.loads(f.read())
        t = cls(
            trainable if trainable is not None else
            cloudpickle.loads(state["trainable"]),
            param_space={},  # variants come from the saved state
            tune_config=state["tune_config"],
            run_config=RunConfig(
                name=os.path.basename(path.rstrip("/")),
                storage_path=os.path.dirname(path.rstrip("/")),
            ),
        )
        t._restored = state
        return t

    def fit(self) -> ResultGrid:
        import cloudpickle

        import ray_amd as ray

        tc = self.tune_config
        storage = self.run_config.resolved_storage_path()
        os.makedirs(storage, exist_ok=True)

        # Trainer-as-trainable (reference: Train trainers are Trainables)
        from ..train.trainer import DataParallelTrainer

        if isinstance(self._trainable, DataParallelTrainer):
            return self._fit_trainer_trials(storage)

        done_names = set()
        restored_results: List[Result] = []
        # search_alg drives sequential ask/tell config proposal
        # (reference: tune/search/ Searcher protocol); without one,
        # variants are pre-generated
        searcher = tc.search_alg if self._restored is None else None
        if searcher is not None:
            searcher.set_search_properties(
                tc.metric, tc.mode, self.param_space)
            variants = []
            n_target = tc.num_samples
        elif self._restored is not None:
            variants = self._restored["variants"]
            n_target = len(variants)
        else:
            variants = generate_variants(self.param_space, tc.num_samples)
            n_target = len(variants)
        fn_bytes = cloudpickle.dumps(self._trainable)
        # persist experiment state so Tuner.restore can resume it
        with open(os.path.join(storage, "tuner_state.pkl"), "wb") as f:
            f.write(cloudpickle.dumps({
                "variants": variants,
                "tune_config": tc,
                "trainable": fn_bytes,
            }))
        if self._restored is not None:
            import json as _json

            for i in range(len(variants)):
                rp = os.path.join(storage, f"trial_{i:05d}", "result.json")
                if os.path.exists(rp):
                    with open(rp) as f:
                        row = _json.load(f)
                    if row.get("error"):
                        continue  # re-run failed trials
                    done_names.add(i)
                    restored_results.append(Result(
                        metrics=row.get("metrics"),
                        checkpoint=(Checkpoint(row["ckpt"])
                                    if row.get("ckpt") else None),
                        path=os.path.join(storage, f"trial_{i:05d}"),
                        error=None,
                    ))
        scheduler = tc.scheduler or FIFOScheduler()
        max_conc = tc.max_concurrent_trials or min(8, max(1, n_target))
        callbacks = list(getattr(self.run_config, "callbacks", None) or [])

        def _cb(method, *a, **k):
            for cb in callbacks:
                try:
                    getattr(cb, method)(*a, **k)
                except Exception:
                    import traceback as _tb

                    _tb.print_exc()

        _cb("setup")
        Actor = ray.remote(_TrialActor)

        trials = []  # dicts: actor, config, rows, done, error, it
        pending = [
            (i, cfg) for i, cfg in enumerate(variants) if i not in done_names
        ]
        running: List[dict] = []
        finished: List[dict] = []
        t_start = time.time()

        def launch(idx_cfg):
            idx, cfg = idx_cfg
            a = Actor.options(num_cpus=1).remote()
            name = f"trial_{idx:05d}"
            # fire-and-forget: the actor may stay PENDING until a CPU
            # frees up; the fetch loop below is fully non-blocking.
            a.run.remote(fn_bytes, cfg, os.path.join(storage, name), name)
            t = {"actor": a, "config": cfg, "rows": [], "done": False,
                 "error": None, "it": 0, "name": name, "stopped": False,
                 "ckpt": None, "fetch_ref": None}
            if scheduler is not None and hasattr(scheduler, "observe_config"):
                scheduler.observe_config(name, cfg)
            running.append(t)
            _cb("on_trial_start", t["it"], running, t)

        drawn = 0
        while pending or running or (searcher is not None
                                     and drawn < n_target):
            while pending and len(running) < max_conc:
                launch(pending.pop(0))
            while (searcher is not None and drawn < n_target
                   and len(running) < max_conc):
                cfg = searcher.suggest(f"trial_{drawn:05d}")
                if cfg is None:
                    break  # searcher-imposed concurrency cap
                variants.append(cfg)
                launch((drawn, cfg))
                drawn += 1
            # issue fetches and wait for any to become ready
            ref_to_trial = {}
            for t in running:
                if t["fetch_ref"] is None:
                    t["fetch_ref"] = t["actor"].fetch.remote()
                ref_to_trial[t["fetch_ref"]] = t
            if not ref_to_trial:
                time.sleep(0.02)
                continue
            ready, _ = ray.wait(
                list(ref_to_trial), num_returns=len(ref_to_trial), timeout=0.2
            )
            for ref in ready:
                t = ref_to_trial[ref]
                t["fetch_ref"] = None
                try:
                    st = ray.get(ref, timeout=30)
                except (ray.exceptions.RayActorError, ray.exceptions.RayError):
                    t["error"] = "trial actor died"
                    t["done"] = True
                    running.remove(t)
                    finished.append(t)
                    if searcher is not None:
                        searcher.on_trial_complete(t["name"], None,
                                                   error=True)
                    continue
                for r in st["results"]:
                    t["it"] += 1
                    row = dict(r["metrics"])
                    row["training_iteration"] = t["it"]
                    row.update(
                        {f"config/{k}": v for k, v in t["config"].items()}
                    )
                    t["rows"].append(row)
                    _cb("on_trial_result", t["it"], running, t, row)
                    if r.get("checkpoint_path"):
                        t["ckpt"] = r["checkpoint_path"]
                    mv = r["metrics"].get(tc.metric) if tc.metric else None
                    decision = scheduler.on_trial_result(t["name"], t["it"], mv)
                    if decision == "STOP" and not t["stopped"]:
                        t["stopped"] = True
                        t["actor"].request_stop.remote()
                if (
                    isinstance(scheduler, PopulationBasedTraining)
                    and not t["stopped"]
                    and t["rows"]
                    and scheduler.should_perturb(t["name"], t["it"])
                ):
                    scheduler._last_perturb[t["name"]] = t["it"]
                    donor_name = scheduler.exploit_target(t["name"])
                    donor = next(
                        (x for x in running if x["name"] == donor_name), None
                    )
                    if donor is not None and donor is not t:
                        self._pbt_exploit(
                            ray, scheduler, t, donor, fn_bytes, storage
                        )
                if st["done"] or (t["stopped"] and not st["results"]):
                    t["error"] = t["error"] or st["error"]
                    t["done"] = True
                    running.remove(t)
                    finished.append(t)
                    if searcher is not None:
                        searcher.on_trial_complete(
                            t["name"],
                            t["rows"][-1] if t["rows"] else None,
                            error=bool(t["error"]),
                        )
                    _cb("on_trial_error" if t["error"]
                        else "on_trial_complete", t["it"], running, t)
                    try:
                        ray.kill(t["actor"])
                    except Exception:
                        pass
            if tc.time_budget_s and time.time() - t_start > tc.time_budget_s:
                for t in running:
                    t["actor"].request_stop.remote()

        _cb("on_experiment_end", finished)
        results = list(restored_results)
        import json as _json

        for t in finished:
            metrics = t["rows"][-1] if t["rows"] else None
            results.append(
                Result(
                    metrics=metrics,
                    checkpoint=Checkpoint(t["ckpt"]) if t["ckpt"] else None,
                    path=os.path.join(storage, t["name"]),
                    error=RuntimeError(t["error"]) if t["error"] else None,
                )
            )
            # per-trial completion record (drives Tuner.restore)
            try:
                with open(os.path.join(storage, t["name"],
                                       "result.json"), "w") as f:
                    _json.dump({
                        "metrics": metrics, "ckpt": t["ckpt"],
                        "error": t["error"],
                    }, f, default=str)
            except OSError:
                pass
        return ResultGrid(results, tc.metric, tc.mode)

    def _pbt_exploit(self, ray, scheduler, t, donor, fn_bytes, storage):
        """Restart trial `t` from `donor`'s latest checkpoint with a
        perturbed clone of the donor's config."""
        try:
            ckpt = ray.get(
                donor["actor"].latest_checkpoint_path.remote(), timeout=15
            )
        except ray.exceptions.RayError:
            ckpt = donor.get("ckpt")
        if ckpt is None:
            ckpt = donor.get("ckpt")
        new_cfg = scheduler.explore(donor["config"])
        if hasattr(scheduler, "observe_config"):
            scheduler.observe_config(t["name"], new_cfg)
        try:
            ray.kill(t["actor"])
        except Exception:
            pass
        a = ray.remote(_TrialActor).options(num_cpus=1).remote()
        a.run.remote(
            fn_bytes, new_cfg, os.path.join(storage, t["name"]), t["name"],
            restore_path=ckpt,
        )
        t["actor"] = a
        t["config"] = new_cfg
        t["fetch_ref"] = None
        if ckpt:
            t["ckpt"] = ckpt
        scheduler.num_perturbations += 1

    def _fit_trainer_trials(self, storage: str) -> ResultGrid:
        tc = self.tune_config
        space = self.param_space.get("train_loop_config", {})
        variants = generate_variants(space, tc.num_samples) if space else [
            {} for _ in range(tc.num_samples)
        ]
        results = []
        for i, cfg in enumerate(variants):
            trainer = self._trainable
            merged = dict(trainer._config)
            merged.update(cfg)
            t2 = type(trainer)(
                trainer._fn,
                train_loop_config=merged,
                scaling_config=trainer.scaling_config,
                run_config=RunConfig(
                    name=f"{self.run_config.name}_t{i}", storage_path=storage
                ),
                datasets=trainer.datasets,
            )
            res = t2.fit()
            if res.metrics is not None:
                res.metrics.update({f"config/{k}": v for k, v in cfg.items()})
            results.append(res)
        return ResultGrid(results, tc.metric, tc.mode)


def report(metrics: Dict[str, Any], checkpoint=None):
    from ..train.session import report as _r

    _r(metrics, checkpoint)


def with_parameters(fn, **params):
    def wrapped(config):
        return fn(config, **params)

    return wrapped


def with_resources(fn, resources):
    fn._tune_resources = resources
    return fn


class Callback:
    """Experiment-lifecycle hooks (reference: tune/callback.py) —
    attach via RunConfig(callbacks=[...])."""

    def setup(self, **info):
        pass

    def on_trial_start(self, iteration, trials, trial, **info):
        pass

    def on_trial_result(self, iteration, trials, trial, result, **info):
        pass

    def on_trial_complete(self, iteration, trials, trial, **info):
        pass

    def on_trial_error(self, iteration, trials, trial, **info):
        pass

    def on_experiment_end(self, trials, **info):
        pass


def run(trainable, config=None, num_samples=1, metric=None, mode="max",
        scheduler=None, **kwargs):
    """Legacy tune.run API shim over Tuner."""
    tuner = Tuner(
        trainable,
        param_space=config or {},
        tune_config=TuneConfig(metric=metric, mode=mode,
                               num_samples=num_samples, scheduler=scheduler),
    )
    return tuner.fit()
