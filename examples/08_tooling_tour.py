"""Tour of the tooling/API surface added in round 2 — all CPU-runnable.

    python examples/08_tooling_tour.py

Covers: Dask-on-Ray graph execution, Dataset expressions + aggregate
pushdown, GP-EI Bayesian hyperparameter search, the sklearn GBDT
trainer, and graceful node drain.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import ray_amd as ray
from ray_amd import tune
from ray_amd.tune import BayesOptSearch, TuneConfig, Tuner

ray.init(num_cpus=4)

# --- Dask-on-Ray: plain dask graphs, one ray task per node ----------
from ray_amd.util.dask import ray_dask_get


def inc(x):
    return x + 1


dsk = {"a": 1, "b": (inc, "a"), "c": (sum, [(inc, "b"), "b", "a"])}
print("dask graph c =", ray_dask_get(dsk, "c"))  # (3)+2+1 = 6

# --- Dataset expressions + pushed-down aggregates -------------------
import ray_amd.data as rd
from ray_amd.data import col, lit

ds = rd.from_items([{"x": i, "y": i % 5} for i in range(100)],
                   parallelism=8)
filtered = ds.filter(expr=(col("x") > 20) & (col("y") == lit(0)))
print("filtered count:", filtered.count())
print("grouped mean:",
      ds.groupby("y").mean("x").take(2))  # map-side combined

# --- Bayesian search over a quadratic -------------------------------
def objective(config):
    tune.report({"score": -(config["lr"] - 0.3) ** 2})


grid = Tuner(
    objective,
    param_space={"lr": tune.uniform(0.0, 1.0)},
    tune_config=TuneConfig(metric="score", mode="max", num_samples=10,
                           search_alg=BayesOptSearch(n_startup_trials=4,
                                                     seed=0)),
).fit()
print("best lr:", round(grid.get_best_result().metrics["config/lr"], 3))

# --- sklearn trainer on a Dataset -----------------------------------
from sklearn.linear_model import LogisticRegression

from ray_amd.train.gbdt import SklearnTrainer

rows = [{"a": float(a), "b": float(b), "label": int(a + b > 1.0)}
        for a, b in np.random.default_rng(0).random((200, 2))]
res = SklearnTrainer(
    estimator=LogisticRegression(),
    datasets={"train": rd.from_items(rows)},
    label_column="label",
).fit()
print("sklearn train score:", round(res.metrics["train_score"], 3))

ray.shutdown()
print("tour done")
