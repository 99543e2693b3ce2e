"""ray_amd.train tests: TorchTrainer DDP on CPU (gloo, world_size=2),
report/checkpoint/resume, failure handling.
Reference model: python/ray/train/tests/."""
import os

import pytest
import torch

import ray_amd as ray
from ray_amd.train import (
    Checkpoint,
    FailureConfig,
    RunConfig,
    ScalingConfig,
)
from ray_amd.train.torch import TorchTrainer


def train_fn_basic(config):
    import ray_amd.train as train

    ctx = train.get_context()
    assert ctx.get_world_size() == 2
    for i in range(3):
        train.report({"loss": 1.0 / (i + 1), "rank": ctx.get_world_rank()})


def test_torch_trainer_reports(ray_start_regular, tmp_path):
    t = TorchTrainer(
        train_fn_basic,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="t1", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is None
    assert res.metrics["loss"] == pytest.approx(1 / 3)
    assert len(res.metrics_dataframe) == 3


def train_fn_ddp(config):
    import torch.distributed as dist

    import ray_amd.train as train
    from ray_amd.train.torch import prepare_model

    model = torch.nn.Linear(4, 2)
    model = prepare_model(model)
    assert isinstance(model, torch.nn.parallel.DistributedDataParallel)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(8, 4)
    y = torch.randn(8, 2)
    for _ in range(2):
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
    # all ranks see identical (averaged) grads -> identical weights
    w = model.module.weight.detach().flatten()
    gathered = [torch.zeros_like(w) for _ in range(2)]
    dist.all_gather(gathered, w)
    assert torch.allclose(gathered[0], gathered[1])
    train.report({"final_loss": float(loss)})


def test_torch_trainer_ddp_gloo(ray_start_regular, tmp_path):
    t = TorchTrainer(
        train_fn_ddp,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="ddp", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is None
    assert "final_loss" in res.metrics


def train_fn_ckpt(config):
    import tempfile

    import ray_amd.train as train

    ctx = train.get_context()
    start = 0
    ckpt = train.get_checkpoint()
    if ckpt is not None:
        with ckpt.as_directory() as d:
            start = int(open(os.path.join(d, "it.txt")).read())
    for i in range(start, start + 2):
        with tempfile.TemporaryDirectory() as d:
            with open(os.path.join(d, "it.txt"), "w") as f:
                f.write(str(i + 1))
            if ctx.get_world_rank() == 0:
                train.report({"it": i + 1}, checkpoint=Checkpoint.from_directory(d))
            else:
                train.report({"it": i + 1})


def test_checkpoint_and_resume(ray_start_regular, tmp_path):
    t = TorchTrainer(
        train_fn_ckpt,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="ck", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is None
    assert res.metrics["it"] == 2
    assert res.checkpoint is not None
    with res.checkpoint.as_directory() as d:
        assert open(os.path.join(d, "it.txt")).read() == "2"

    # resume
    t2 = TorchTrainer(
        train_fn_ckpt,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="ck2", storage_path=str(tmp_path)),
        resume_from_checkpoint=res.checkpoint,
    )
    res2 = t2.fit()
    assert res2.metrics["it"] == 4


_fail_flag = os.path.join("/tmp", "ray_amd_train_fail_once")


def train_fn_flaky(config):
    import ray_amd.train as train

    if not os.path.exists(config["flag"]):
        open(config["flag"], "w").close()
        raise RuntimeError("injected failure")
    train.report({"ok": 1})


def test_failure_retry(ray_start_regular, tmp_path):
    flag = str(tmp_path / "flag")
    t = TorchTrainer(
        train_fn_flaky,
        train_loop_config={"flag": flag},
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(
            name="flaky", storage_path=str(tmp_path),
            failure_config=FailureConfig(max_failures=1),
        ),
    )
    res = t.fit()
    assert res.error is None
    assert res.metrics["ok"] == 1


def test_failure_no_retry(ray_start_regular, tmp_path):
    def always_fail(config):
        raise RuntimeError("always fails")

    t = TorchTrainer(
        always_fail,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="nf", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is not None
    assert "always fails" in str(res.error)


def test_checkpoint_num_to_keep(ray_start_regular, tmp_path):
    import os

    from ray_amd.train import CheckpointConfig

    def loop(config):
        import tempfile

        import ray_amd.train as train

        for i in range(5):
            with tempfile.TemporaryDirectory() as d:
                open(os.path.join(d, "x.txt"), "w").write(str(i))
                train.report({"i": i}, checkpoint=Checkpoint.from_directory(d))

    t = TorchTrainer(
        loop,
        scaling_config=ScalingConfig(num_workers=1),
        run_config=RunConfig(
            name="keep", storage_path=str(tmp_path),
            checkpoint_config=CheckpointConfig(num_to_keep=2),
        ),
    )
    res = t.fit()
    assert res.error is None
    run_dir = os.path.join(str(tmp_path), "keep")
    ckpts = [d for d in os.listdir(run_dir) if d.startswith("checkpoint_")]
    assert len(ckpts) <= 2


def test_elastic_restart_with_fewer_workers(ray_start_cluster, tmp_path):
    """Elastic policy: after losing a node, the retry restarts with the
    workers that still fit (reference: v2 elastic scaling policy)."""
    import time

    cluster = ray_start_cluster  # head 4 CPUs
    extra = cluster.add_node(num_cpus=2, resources={"extra": 2})
    cluster.connect()
    cluster.wait_for_nodes()
    from ray_amd.train import FailureConfig, RunConfig, ScalingConfig
    from ray_amd.train.torch import TorchTrainer

    flag = str(tmp_path / "fail_once")

    def loop(config):
        import os

        import ray_amd.train as train

        import time as _t

        ctx = train.get_context()
        if not os.path.exists(config["flag"]) and ctx.get_world_rank() == 0:
            open(config["flag"], "w").close()
            _t.sleep(1.5)  # let the chopper remove the extra node first
            raise RuntimeError("injected failure while node shrinks")
        train.report({"world": ctx.get_world_size()})

    # 6 single-CPU workers fit across both nodes initially
    t = TorchTrainer(
        loop,
        train_loop_config={"flag": flag},
        scaling_config=ScalingConfig(num_workers=5, elastic=True,
                                     min_workers=1),
        run_config=RunConfig(
            name="elastic", storage_path=str(tmp_path),
            failure_config=FailureConfig(max_failures=2),
        ),
    )
    # remove the extra node while the first attempt fails
    import threading

    def chopper():
        time.sleep(0.5)
        cluster.remove_node(extra)

    threading.Thread(target=chopper, daemon=True).start()
    res = t.fit()
    assert res.error is None
    assert res.metrics["world"] <= 4  # shrunk below the original 5


def test_huggingface_model_in_trainer(ray_start_regular, tmp_path):
    """HF Transformers models train inside TorchTrainer (reference:
    train/huggingface integration). Random-init config (offline image)."""

    def loop(config):
        import torch
        from transformers import BertConfig, BertForSequenceClassification

        import ray_amd.train as train
        from ray_amd.train.torch import prepare_model

        cfg = BertConfig(
            vocab_size=128, hidden_size=32, num_hidden_layers=2,
            num_attention_heads=2, intermediate_size=64,
            max_position_embeddings=64, num_labels=2,
        )
        model = prepare_model(BertForSequenceClassification(cfg))
        opt = torch.optim.AdamW(model.parameters(), lr=5e-4)
        x = torch.randint(0, 128, (8, 16))
        y = torch.randint(0, 2, (8,))
        first = None
        for _ in range(4):
            out = model(input_ids=x, labels=y)
            out.loss.backward()
            opt.step()
            opt.zero_grad()
            if first is None:
                first = float(out.loss)
        train.report({"first": first, "last": float(out.loss)})

    t = TorchTrainer(
        loop,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(name="hf", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is None
    assert res.metrics["last"] <= res.metrics["first"] + 0.1


def test_hf_transformers_trainer_integration(ray_start_regular, tmp_path):
    """transformers.Trainer inside a TorchTrainer worker: HF picks up
    the worker's RANK/WORLD_SIZE env, metrics and checkpoints flow back
    through ray_amd.train.report (reference:
    train/huggingface/transformers prepare_trainer +
    RayTrainReportCallback)."""
    import ray_amd.train as train
    from ray_amd.train import ScalingConfig, RunConfig
    from ray_amd.train.torch import TorchTrainer

    out_dir = str(tmp_path / "hf_out")

    def loop(cfg):
        import torch
        import numpy as np
        from transformers import (Trainer, TrainingArguments)
        from ray_amd.train.huggingface import (RayTrainReportCallback,
                                               prepare_trainer)

        class TinyModel(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.lin = torch.nn.Linear(4, 2)

            def forward(self, x=None, labels=None):
                logits = self.lin(x)
                loss = torch.nn.functional.cross_entropy(logits, labels)
                return {"loss": loss, "logits": logits}

        class DS(torch.utils.data.Dataset):
            def __len__(self):
                return 32

            def __getitem__(self, i):
                return {"x": torch.randn(4), "labels": i % 2}

        args = TrainingArguments(
            output_dir=cfg["out_dir"], num_train_epochs=1,
            per_device_train_batch_size=8, logging_steps=1,
            save_steps=2, save_total_limit=1, report_to=[],
            use_cpu=True, disable_tqdm=True,
        )
        trainer = Trainer(model=TinyModel(), args=args,
                          train_dataset=DS())
        trainer.add_callback(RayTrainReportCallback().unwrap())
        trainer = prepare_trainer(trainer)
        trainer.train()

    t = TorchTrainer(
        loop, train_loop_config={"out_dir": out_dir},
        scaling_config=ScalingConfig(num_workers=1),
        run_config=RunConfig(name="hf_test"),
    )
    result = t.fit()
    assert result.error is None, result.error
    assert result.metrics and "loss" in result.metrics or "step" in (result.metrics or {})


def test_sklearn_trainer(ray_start_regular, tmp_path):
    """SklearnTrainer fits a real estimator in a ray worker and
    checkpoints it (reference: train/sklearn/sklearn_trainer.py)."""
    import numpy as np
    from sklearn.ensemble import HistGradientBoostingClassifier

    import ray_amd.data as rd
    from ray_amd.train import RunConfig
    from ray_amd.train.gbdt import SklearnTrainer

    rng = np.random.default_rng(0)
    rows = [{"a": float(x), "b": float(y),
             "label": int(x + y > 1.0)}
            for x, y in rng.random((200, 2))]
    ds = rd.from_items(rows)
    trainer = SklearnTrainer(
        estimator=HistGradientBoostingClassifier(max_iter=20),
        datasets={"train": ds, "valid": ds},
        label_column="label",
        run_config=RunConfig(name="skl", storage_path=str(tmp_path)),
    )
    res = trainer.fit()
    assert res.error is None, res.error
    assert res.metrics["train_score"] > 0.9
    assert "valid_score" in res.metrics
    model = SklearnTrainer.get_model(res.checkpoint)
    import pandas as pd

    pred = model.predict(pd.DataFrame([{"a": 0.9, "b": 0.9}]))
    assert pred[0] == 1


def test_xgboost_trainer_missing_lib(ray_start_regular, tmp_path):
    import sys

    if "xgboost" in sys.modules:
        pytest.skip("xgboost installed")
    import ray_amd.data as rd
    from ray_amd.train import RunConfig
    from ray_amd.train.gbdt import XGBoostTrainer

    t = XGBoostTrainer(
        params={"objective": "binary:logistic"},
        datasets={"train": rd.from_items([{"x": 1.0, "label": 0}])},
        label_column="label",
        run_config=RunConfig(name="xgb", storage_path=str(tmp_path)),
    )
    res = t.fit()
    assert res.error is not None
    assert "xgboost" in str(res.error)
