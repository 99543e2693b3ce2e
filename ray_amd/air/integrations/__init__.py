"""Experiment-tracker integrations (reference: python/ray/air/
integrations/{wandb,mlflow}.py). The tracker libraries are optional —
imports inside the callbacks are lazy."""
