"""HF Transformers integration (reference:
python/ray/train/huggingface/transformers/_transformers_utils.py —
prepare_trainer + RayTrainReportCallback).

Run a transformers.Trainer inside a TorchTrainer train loop: the
worker's process group (set up by ray_amd.train.trainer.setup_dist via
RANK/WORLD_SIZE/LOCAL_RANK env) is exactly what HF's TrainingArguments
picks up, so distributed data parallelism flows through our RCCL
process group automatically.
"""
from __future__ import annotations

import os
import tempfile
from typing import Optional


def prepare_trainer(trainer):
    """Make an HF Trainer cooperate with the surrounding ray_amd train
    loop: disable its own world-size probing surprises and return it.
    (HF reads RANK/WORLD_SIZE/LOCAL_RANK from the env, which the train
    worker already set.)"""
    args = trainer.args
    # report through ray_amd.train, not HF's own integrations
    try:
        args.report_to = []
    except Exception:
        pass
    # HF uses LOCAL_RANK for device placement; our workers pin devices
    # through HIP_VISIBLE_DEVICES + LOCAL_RANK already.
    if os.environ.get("WORLD_SIZE") and int(os.environ["WORLD_SIZE"]) > 1:
        args.local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    return trainer


class RayTrainReportCallback:
    """transformers.TrainerCallback that forwards HF logs and
    checkpoints to ray_amd.train.report (reference:
    RayTrainReportCallback)."""

    def __init__(self):
        # subclass TrainerCallback lazily so importing this module
        # works without transformers installed
        from transformers import TrainerCallback

        class _CB(TrainerCallback):
            def on_log(cb, args, state, control, logs=None, **kw):
                from ray_amd import train

                if logs and state.is_world_process_zero:
                    metrics = {k: v for k, v in logs.items()
                               if isinstance(v, (int, float))}
                    metrics["step"] = state.global_step
                    metrics["epoch"] = state.epoch or 0
                    try:
                        train.report(metrics)
                    except Exception:
                        pass

            def on_save(cb, args, state, control, **kw):
                from ray_amd import train
                from ray_amd.train import Checkpoint

                if not state.is_world_process_zero:
                    return
                ckpt_dir = os.path.join(
                    args.output_dir, f"checkpoint-{state.global_step}"
                )
                if os.path.isdir(ckpt_dir):
                    try:
                        train.report(
                            {"step": state.global_step},
                            checkpoint=Checkpoint(ckpt_dir),
                        )
                    except Exception:
                        pass

        self._cb = _CB()

    def __getattr__(self, name):
        return getattr(self._cb, name)

    # transformers accepts callback INSTANCES; hand it the real one
    def unwrap(self):
        return self._cb
