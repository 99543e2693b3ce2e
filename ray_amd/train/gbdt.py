"""GBDT / sklearn trainers (reference: python/ray/train/
xgboost/xgboost_trainer.py, lightgbm/lightgbm_trainer.py,
sklearn/sklearn_trainer.py).

SklearnTrainer runs for real (scikit-learn ships in this image);
XGBoostTrainer/LightGBMTrainer bind lazily to their optional
libraries. Training runs in a ray worker sized by ScalingConfig, the
fitted model lands in a Checkpoint, and fit() returns the standard
Result.
"""
from __future__ import annotations

import os
import pickle
import tempfile
from typing import Dict, Optional

from .checkpoint import Checkpoint
from .config import Result, RunConfig, ScalingConfig

MODEL_FILENAME = "model.pkl"
BOOSTER_FILENAME = "model.ubj"


def _ray():
    import ray_amd

    return ray_amd


class _GbdtTrainerBase:
    def __init__(self, *, datasets: Dict, label_column: str,
                 scaling_config: Optional[ScalingConfig] = None,
                 run_config: Optional[RunConfig] = None):
        self.datasets = datasets
        self.label_column = label_column
        self.scaling_config = scaling_config or ScalingConfig()
        self.run_config = run_config or RunConfig(name=type(self).__name__)

    def _storage(self) -> str:
        p = self.run_config.resolved_storage_path()
        os.makedirs(p, exist_ok=True)
        return p

    def _to_xy(self, ds):
        df = ds.to_pandas()
        y = df.pop(self.label_column)
        return df, y

    def fit(self) -> Result:
        ray = _ray()
        storage = self._storage()
        res = self.scaling_config.worker_resources()
        fit_remote = ray.remote(self._fit_worker).options(
            num_cpus=res.get("CPU", 1), num_gpus=res.get("GPU", 0))
        try:
            metrics, ckpt_dir = ray.get(
                fit_remote.remote(self, storage), timeout=3600)
        except Exception as e:  # surface the worker failure as Result
            return Result(metrics=None, checkpoint=None, path=storage,
                          error=e)
        return Result(metrics=metrics, checkpoint=Checkpoint(ckpt_dir),
                      path=storage, error=None)

    # implemented by subclasses (runs inside the worker)
    @staticmethod
    def _fit_worker(self, storage):
        raise NotImplementedError


class SklearnTrainer(_GbdtTrainerBase):
    """Fit any scikit-learn estimator on a Dataset (reference:
    train/sklearn/sklearn_trainer.py)."""

    def __init__(self, *, estimator, datasets, label_column,
                 scaling_config=None, run_config=None):
        super().__init__(datasets=datasets, label_column=label_column,
                         scaling_config=scaling_config,
                         run_config=run_config)
        self.estimator = estimator

    @staticmethod
    def _fit_worker(self, storage):
        X, y = self._to_xy(self.datasets["train"])
        est = self.estimator
        est.fit(X, y)
        metrics = {"train_score": float(est.score(X, y))}
        if "valid" in self.datasets:
            Xv, yv = self._to_xy(self.datasets["valid"])
            metrics["valid_score"] = float(est.score(Xv, yv))
        ckpt_dir = tempfile.mkdtemp(prefix="ckpt_", dir=storage)
        with open(os.path.join(ckpt_dir, MODEL_FILENAME), "wb") as f:
            pickle.dump(est, f)
        return metrics, ckpt_dir

    @staticmethod
    def get_model(checkpoint: Checkpoint):
        with open(os.path.join(checkpoint.path, MODEL_FILENAME),
                  "rb") as f:
            return pickle.load(f)


class XGBoostTrainer(_GbdtTrainerBase):
    """xgboost.train over Dataset shards (reference:
    train/xgboost/xgboost_trainer.py). Requires `xgboost`."""

    def __init__(self, *, params: dict, datasets, label_column,
                 num_boost_round: int = 10, scaling_config=None,
                 run_config=None):
        super().__init__(datasets=datasets, label_column=label_column,
                         scaling_config=scaling_config,
                         run_config=run_config)
        self.params = dict(params)
        self.num_boost_round = num_boost_round

    @staticmethod
    def _fit_worker(self, storage):
        try:
            import xgboost as xgb
        except ImportError as e:
            raise ImportError(
                "XGBoostTrainer requires the `xgboost` package") from e
        X, y = self._to_xy(self.datasets["train"])
        dtrain = xgb.DMatrix(X, label=y)
        evals = [(dtrain, "train")]
        if "valid" in self.datasets:
            Xv, yv = self._to_xy(self.datasets["valid"])
            evals.append((xgb.DMatrix(Xv, label=yv), "valid"))
        evals_result: dict = {}
        booster = xgb.train(self.params, dtrain,
                            num_boost_round=self.num_boost_round,
                            evals=evals, evals_result=evals_result)
        metrics = {f"{split}-{m}": float(v[-1])
                   for split, ms in evals_result.items()
                   for m, v in ms.items()}
        ckpt_dir = tempfile.mkdtemp(prefix="ckpt_", dir=storage)
        booster.save_model(os.path.join(ckpt_dir, BOOSTER_FILENAME))
        return metrics, ckpt_dir

    @staticmethod
    def get_model(checkpoint: Checkpoint):
        import xgboost as xgb

        booster = xgb.Booster()
        booster.load_model(os.path.join(checkpoint.path,
                                        BOOSTER_FILENAME))
        return booster


class LightGBMTrainer(_GbdtTrainerBase):
    """lightgbm.train over Dataset shards (reference:
    train/lightgbm/lightgbm_trainer.py). Requires `lightgbm`."""

    def __init__(self, *, params: dict, datasets, label_column,
                 num_boost_round: int = 10, scaling_config=None,
                 run_config=None):
        super().__init__(datasets=datasets, label_column=label_column,
                         scaling_config=scaling_config,
                         run_config=run_config)
        self.params = dict(params)
        self.num_boost_round = num_boost_round

    @staticmethod
    def _fit_worker(self, storage):
        try:
            import lightgbm as lgb
        except ImportError as e:
            raise ImportError(
                "LightGBMTrainer requires the `lightgbm` package") from e
        X, y = self._to_xy(self.datasets["train"])
        dtrain = lgb.Dataset(X, label=y)
        valid_sets = [dtrain]
        valid_names = ["train"]
        if "valid" in self.datasets:
            Xv, yv = self._to_xy(self.datasets["valid"])
            valid_sets.append(lgb.Dataset(Xv, label=yv))
            valid_names.append("valid")
        record: dict = {}
        booster = lgb.train(
            self.params, dtrain, num_boost_round=self.num_boost_round,
            valid_sets=valid_sets, valid_names=valid_names,
            callbacks=[lgb.record_evaluation(record)])
        metrics = {f"{split}-{m}": float(v[-1])
                   for split, ms in record.items()
                   for m, v in ms.items()}
        ckpt_dir = tempfile.mkdtemp(prefix="ckpt_", dir=storage)
        booster.save_model(os.path.join(ckpt_dir, "model.txt"))
        return metrics, ckpt_dir

    @staticmethod
    def get_model(checkpoint: Checkpoint):
        import lightgbm as lgb

        return lgb.Booster(
            model_file=os.path.join(checkpoint.path, "model.txt"))
