"""Preprocessors (reference: python/ray/data/preprocessors/ — scalers,
encoders, concatenator, torch). fit/transform over Datasets; the torch
normalizer rides the HIP img_normalize kernel on GPU batches."""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np


class Preprocessor:
    _fitted = False

    def fit(self, ds) -> "Preprocessor":
        self._fit(ds)
        self._fitted = True
        return self

    def transform(self, ds):
        return ds.map_batches(self._transform_numpy, batch_format="numpy")

    def fit_transform(self, ds):
        return self.fit(ds).transform(ds)

    def transform_batch(self, batch):
        return self._transform_numpy(batch)

    def _fit(self, ds):
        pass

    def _transform_numpy(self, batch):
        raise NotImplementedError


class StandardScaler(Preprocessor):
    def __init__(self, columns: List[str]):
        self.columns = columns
        self.stats_: Dict[str, tuple] = {}

    def _fit(self, ds):
        for c in self.columns:
            vals = np.concatenate(
                [b[c] for b in ds.iter_batches(batch_size=None, batch_format="numpy")]
            )
            self.stats_[c] = (float(np.mean(vals)), float(np.std(vals)) or 1.0)

    def _transform_numpy(self, batch):
        batch = dict(batch)
        for c in self.columns:
            m, s = self.stats_[c]
            batch[c] = (batch[c] - m) / s
        return batch


class MinMaxScaler(Preprocessor):
    def __init__(self, columns: List[str]):
        self.columns = columns
        self.stats_: Dict[str, tuple] = {}

    def _fit(self, ds):
        for c in self.columns:
            vals = np.concatenate(
                [b[c] for b in ds.iter_batches(batch_size=None, batch_format="numpy")]
            )
            lo, hi = float(np.min(vals)), float(np.max(vals))
            self.stats_[c] = (lo, (hi - lo) or 1.0)

    def _transform_numpy(self, batch):
        batch = dict(batch)
        for c in self.columns:
            lo, rng = self.stats_[c]
            batch[c] = (batch[c] - lo) / rng
        return batch


class LabelEncoder(Preprocessor):
    def __init__(self, label_column: str):
        self.label_column = label_column
        self.classes_: Optional[np.ndarray] = None

    def _fit(self, ds):
        vals = np.concatenate(
            [
                b[self.label_column]
                for b in ds.iter_batches(batch_size=None, batch_format="numpy")
            ]
        )
        self.classes_ = np.unique(vals)

    def _transform_numpy(self, batch):
        batch = dict(batch)
        idx = np.searchsorted(self.classes_, batch[self.label_column])
        batch[self.label_column] = idx.astype(np.int64)
        return batch


class Concatenator(Preprocessor):
    def __init__(self, columns: List[str], output_column_name: str = "concat_out",
                 dtype=None, drop: bool = True):
        self.columns = columns
        self.output_column_name = output_column_name
        self.dtype = dtype
        self.drop = drop
        self._fitted = True

    def _transform_numpy(self, batch):
        batch = dict(batch)
        cols = [np.atleast_2d(batch[c].reshape(len(batch[c]), -1))
                for c in self.columns]
        out = np.concatenate(cols, axis=1)
        if self.dtype is not None:
            out = out.astype(self.dtype)
        if self.drop:
            for c in self.columns:
                del batch[c]
        batch[self.output_column_name] = out
        return batch


class TorchVisionNormalizer(Preprocessor):
    """uint8 NHWC image batches -> normalized bf16 NCHW tensors.
    On GPU uses the fused HIP cast+normalize kernel
    (ray_amd/csrc/hip/elementwise.hip img_norm_u8_bf16)."""

    def __init__(self, column: str = "image",
                 mean=(0.485, 0.456, 0.406), std=(0.229, 0.224, 0.225)):
        self.column = column
        self.mean = np.asarray(mean, dtype=np.float32)
        self.std = np.asarray(std, dtype=np.float32)
        self._fitted = True

    def _transform_numpy(self, batch):
        batch = dict(batch)
        x = batch[self.column]
        xf = x.astype(np.float32) / 255.0
        xf = (xf - self.mean) / self.std
        batch[self.column] = np.moveaxis(xf, -1, 1)
        return batch

    def transform_torch_gpu(self, x_u8):
        import torch

        from ray_amd import ops

        mean = torch.as_tensor(self.mean, device=x_u8.device)
        std = torch.as_tensor(self.std, device=x_u8.device)
        return ops.img_normalize(x_u8, mean, std)
