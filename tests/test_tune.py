"""Tune tests (reference model: python/ray/tune/tests/)."""
import pytest

import ray_amd as ray
from ray_amd import tune
from ray_amd.tune import ASHAScheduler, TuneConfig, Tuner


def trainable_quadratic(config):
    # minimize (x-3)^2 reported over a few iterations
    x = config["x"]
    for i in range(3):
        tune.report({"score": -((x - 3.0) ** 2), "x": x})


def test_grid_search(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig

    tuner = Tuner(
        trainable_quadratic,
        param_space={"x": tune.grid_search([0.0, 1.0, 3.0, 5.0])},
        tune_config=TuneConfig(metric="score", mode="max"),
        run_config=RunConfig(name="grid", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 4
    best = grid.get_best_result()
    assert best.metrics["x"] == 3.0


def test_random_search_uniform(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig

    tuner = Tuner(
        trainable_quadratic,
        param_space={"x": tune.uniform(0, 6)},
        tune_config=TuneConfig(metric="score", mode="max", num_samples=6),
        run_config=RunConfig(name="rand", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 6
    assert all(0 <= r.metrics["x"] <= 6 for r in grid if r.metrics)


def test_variant_generation():
    from ray_amd.tune.impl import generate_variants

    vs = generate_variants(
        {"a": tune.grid_search([1, 2]), "b": tune.choice([10]), "c": 5},
        num_samples=2,
    )
    assert len(vs) == 4
    assert all(v["c"] == 5 and v["b"] == 10 for v in vs)
    assert sorted(v["a"] for v in vs) == [1, 1, 2, 2]


def test_asha_stops_bad_trials(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig

    def slow_trainable(config):
        import time

        # slow enough that trials overlap even when starts stagger
        # under full-suite load (ASHA needs rung comparisons)
        for i in range(15):
            tune.report({"score": config["x"] * (i + 1)})
            time.sleep(0.1)

    sched = ASHAScheduler(metric="score", mode="max", max_t=15,
                          grace_period=2, reduction_factor=2)
    tuner = Tuner(
        slow_trainable,
        param_space={"x": tune.grid_search([0.1, 0.2, 1.0, 2.0])},
        tune_config=TuneConfig(metric="score", mode="max", scheduler=sched,
                               max_concurrent_trials=4),
        run_config=RunConfig(name="asha", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    best = grid.get_best_result()
    assert best.metrics["config/x"] == 2.0
    # at least one weak trial should have been stopped early
    iters = [r.metrics["training_iteration"] for r in grid if r.metrics]
    assert min(iters) < 15


def test_trial_error_captured(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig

    def failing(config):
        if config["x"] == 1:
            raise ValueError("bad trial")
        tune.report({"ok": 1})

    tuner = Tuner(
        failing,
        param_space={"x": tune.grid_search([0, 1])},
        tune_config=TuneConfig(metric="ok", mode="max"),
        run_config=RunConfig(name="err", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid.errors) == 1
    best = grid.get_best_result()
    assert best.metrics["ok"] == 1


def test_tuner_with_torch_trainer(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig, ScalingConfig
    from ray_amd.train.torch import TorchTrainer

    def loop(config):
        import ray_amd.train as train

        train.report({"lr_used": config["lr"]})

    trainer = TorchTrainer(
        loop,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="base", storage_path=str(tmp_path)),
    )
    tuner = Tuner(
        trainer,
        param_space={"train_loop_config": {"lr": tune.grid_search([0.1, 0.2])}},
        tune_config=TuneConfig(metric="lr_used", mode="max"),
        run_config=RunConfig(name="tt", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 2
    assert grid.get_best_result().metrics["lr_used"] == 0.2


def test_pbt_exploit_and_explore(ray_start_regular, tmp_path):
    """Bottom-quantile trials adopt a top trial's checkpoint + perturbed
    config (reference: tune/schedulers/pbt.py)."""
    from ray_amd import tune

    def trainable(config):
        import json
        import os as _os
        import time as _tm

        ckpt = tune.get_checkpoint()
        score = 0.0
        if ckpt is not None:
            with open(_os.path.join(ckpt.path, "state.json")) as f:
                score = json.load(f)["score"]
        for _ in range(30):
            score += config["lr"]  # higher lr -> faster "learning"
            d = str(tmp_path / f"ck_{_os.getpid()}_{_tm.time_ns()}")
            _os.makedirs(d, exist_ok=True)
            with open(_os.path.join(d, "state.json"), "w") as f:
                json.dump({"score": score}, f)
            tune.report({"score": score},
                        checkpoint=tune.Checkpoint(d))
            _tm.sleep(0.05)

    sched = tune.PopulationBasedTraining(
        metric="score", mode="max", perturbation_interval=5,
        hyperparam_mutations={"lr": tune.uniform(0.1, 1.0)}, seed=1,
    )
    tuner = tune.Tuner(
        trainable,
        param_space={"lr": tune.grid_search([0.01, 0.02, 0.9, 1.0])},
        tune_config=tune.TuneConfig(metric="score", mode="max",
                                    scheduler=sched,
                                    max_concurrent_trials=4),
    )
    rg = tuner.fit()
    assert sched.num_perturbations >= 1
    best = rg.get_best_result()
    assert best.metrics["score"] > 5.0


def test_median_stopping_rule(ray_start_regular):
    from ray_amd import tune

    def trainable(config):
        import time as _tm

        for i in range(50):
            tune.report({"score": config["base"] + i})
            _tm.sleep(0.05)

    sched = tune.MedianStoppingRule(metric="score", mode="max",
                                    grace_period=5)
    rg = tune.Tuner(
        trainable,
        param_space={
            "base": tune.grid_search([0.0, 0.0, 100.0, 100.0]),
        },
        tune_config=tune.TuneConfig(metric="score", mode="max",
                                    scheduler=sched,
                                    max_concurrent_trials=4),
    ).fit()
    iters = sorted(r.metrics["training_iteration"] for r in rg
                   if r.metrics)
    # low-base trials get median-stopped before the high ones finish
    assert iters[0] < 50, iters
    best = rg.get_best_result()
    assert best.metrics["score"] >= 100


def test_tuner_restore_resumes_unfinished(ray_start_regular, tmp_path):
    """Tuner.restore re-runs only failed/missing trials; finished ones
    load from disk (reference: Tuner.restore)."""
    from ray_amd.train import RunConfig

    flag = tmp_path / "fixed"
    runs = tmp_path / "runs"
    runs.mkdir()

    def trainable(config, flag=str(flag), runs=str(runs)):
        import os
        import time as _tm

        with open(os.path.join(runs, f"run_{config['x']}_{_tm.time_ns()}"),
                  "w") as f:
            f.write("x")
        if config["x"] == 3 and not os.path.exists(flag):
            raise RuntimeError("transient failure")
        tune.report({"score": config["x"] * 10})

    space = {"x": tune.grid_search([1, 2, 3, 4])}
    exp_dir = str(tmp_path / "exp")
    tuner = Tuner(
        trainable, param_space=space,
        tune_config=TuneConfig(metric="score", mode="max"),
        run_config=RunConfig(name="exp", storage_path=str(tmp_path)),
    )
    rg = tuner.fit()
    assert len(rg.errors) == 1
    n_first = len(list(runs.iterdir()))
    assert n_first == 4

    flag.write_text("ok")
    rg2 = Tuner.restore(exp_dir).fit()
    assert len(rg2.errors) == 0
    assert sorted(r.metrics["score"] for r in rg2 if r.metrics) == [
        10, 20, 30, 40
    ]
    # only the failed trial re-ran
    assert len(list(runs.iterdir())) == n_first + 1


def test_pb2_scheduler(ray_start_regular, tmp_path):
    """PB2 (reference: tune/schedulers/pb2.py): GP-UCB explore over
    hyperparam_bounds replaces random perturbation; bottom-quantile
    trials restart from top-quantile checkpoints with GP-chosen
    configs."""
    from ray_amd import tune
    from ray_amd.tune import PB2

    def trainable(config):
        import os

        for it in range(8):
            # score peaks at lr=0.5
            score = -abs(config["lr"] - 0.5) * 10 + it * 0.1
            ckpt_dir = str(tmp_path / f"ck_{os.getpid()}_{it}")
            os.makedirs(ckpt_dir, exist_ok=True)
            with open(os.path.join(ckpt_dir, "s.txt"), "w") as f:
                f.write(str(it))
            from ray_amd.train import Checkpoint

            tune.report({"score": score},
                        checkpoint=Checkpoint(ckpt_dir))

    sched = PB2(metric="score", mode="max", perturbation_interval=2,
                hyperparam_bounds={"lr": [0.0, 1.0]})
    tuner = tune.Tuner(
        trainable,
        param_space={"lr": tune.uniform(0.0, 1.0)},
        tune_config=tune.TuneConfig(metric="score", mode="max",
                                    num_samples=4, scheduler=sched),
    )
    results = tuner.fit()
    best = results.get_best_result()
    assert best.metrics["score"] is not None


def test_bayesopt_search_end_to_end(ray_start_regular, tmp_path):
    """Native GP-EI searcher (reference: search/bayesopt): sequential
    ask/tell through the Tuner; later suggests cluster near the
    optimum of the quadratic."""
    from ray_amd.train import RunConfig
    from ray_amd.tune import BayesOptSearch

    search = BayesOptSearch(n_startup_trials=6, seed=7)
    tuner = Tuner(
        trainable_quadratic,
        param_space={"x": tune.uniform(0, 6)},
        tune_config=TuneConfig(metric="score", mode="max", num_samples=14,
                               search_alg=search,
                               max_concurrent_trials=2),
        run_config=RunConfig(name="bo", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 14
    best = grid.get_best_result()
    # GP-EI should land close to x=3 after the random startup phase
    assert abs(best.metrics["x"] - 3.0) < 1.0, best.metrics


def test_bayesopt_ask_tell_unit():
    """The GP phase proposes near the optimum once observations
    bracket it (no cluster needed)."""
    from ray_amd.tune import BayesOptSearch

    s = BayesOptSearch(metric="score", mode="max", n_startup_trials=5,
                       seed=3)
    s.set_search_properties("score", "max", {"x": tune.uniform(0.0, 1.0)})
    for i in range(12):
        cfg = s.suggest(f"t{i}")
        s.on_trial_complete(
            f"t{i}", {"score": -((cfg["x"] - 0.7) ** 2)})
    xs = [s._decode(u)["x"] for u in (s._ei_argmax(),)]
    assert abs(xs[0] - 0.7) < 0.2


def test_concurrency_limiter():
    from ray_amd.tune import BasicVariantGenerator, ConcurrencyLimiter

    base = BasicVariantGenerator()
    base.set_search_properties(None, "max", {"x": tune.uniform(0, 1)})
    lim = ConcurrencyLimiter(base, max_concurrent=2)
    a = lim.suggest("a")
    b = lim.suggest("b")
    assert a is not None and b is not None
    assert lim.suggest("c") is None  # capped
    lim.on_trial_complete("a", {"x": 1})
    assert lim.suggest("c") is not None


def test_optuna_search_adapter(monkeypatch):
    """Adapter glue against a minimal fake optuna (library optional in
    this image): space conversion + ask/tell routing."""
    import sys
    import types

    told = []

    class FakeTrial:
        def __init__(self, n):
            self.n = n

        def suggest_float(self, k, lo, hi, log=False):
            return lo + 0.5 * (hi - lo)

        def suggest_int(self, k, lo, hi):
            return lo

        def suggest_categorical(self, k, options):
            return options[0]

    class FakeStudy:
        def __init__(self):
            self._n = 0

        def ask(self):
            t = FakeTrial(self._n)
            self._n += 1
            return t

        def tell(self, trial, value=None, state=None):
            told.append((trial.n, value, state))

    fake = types.ModuleType("optuna")
    fake.samplers = types.SimpleNamespace(TPESampler=lambda seed=None: None)
    fake.create_study = lambda sampler=None, direction=None: FakeStudy()
    fake.trial = types.SimpleNamespace(
        TrialState=types.SimpleNamespace(FAIL="FAIL"))
    monkeypatch.setitem(sys.modules, "optuna", fake)

    from ray_amd.tune.search import OptunaSearch

    s = OptunaSearch(metric="score", mode="max")
    s.set_search_properties("score", "max", {
        "lr": tune.loguniform(1e-4, 1e-1),
        "layers": tune.choice([2, 4]),
        "fixed": 7,
    })
    cfg = s.suggest("t0")
    assert cfg["fixed"] == 7 and cfg["layers"] == 2
    assert 1e-4 <= cfg["lr"] <= 1e-1
    s.on_trial_complete("t0", {"score": 1.5})
    assert told == [(0, 1.5, None)]
    s.suggest("t1")
    s.on_trial_complete("t1", None, error=True)
    assert told[-1] == (1, None, "FAIL")


def test_optuna_missing_raises():
    import builtins
    import sys

    if "optuna" in sys.modules:
        pytest.skip("optuna installed")
    from ray_amd.tune import OptunaSearch

    with pytest.raises(ImportError, match="optuna"):
        OptunaSearch()


def test_hyperband_scheduler(ray_start_regular, tmp_path):
    """HyperBand brackets (reference: schedulers/hyperband.py): bad
    trials stop at rung boundaries, good ones run to max_t; brackets
    get different grace periods."""
    from ray_amd.train import RunConfig
    from ray_amd.tune import HyperBandScheduler

    def trainable(config):
        for i in range(9):
            tune.report({"score": config["x"] + i * 0.01})

    sched = HyperBandScheduler(metric="score", mode="max", max_t=9,
                               reduction_factor=3)
    tuner = Tuner(
        trainable,
        param_space={"x": tune.grid_search([0.0, 1.0, 2.0, 3.0, 4.0,
                                            5.0])},
        tune_config=TuneConfig(metric="score", mode="max",
                               scheduler=sched,
                               max_concurrent_trials=3),
        run_config=RunConfig(name="hb", storage_path=str(tmp_path)),
    )
    grid = tuner.fit()
    assert len(grid) == 6
    best = grid.get_best_result()
    assert best.metrics["config/x"] == 5.0
    # brackets were assigned round-robin with distinct grace periods
    assert len(set(sched._bracket_of.values())) > 1
